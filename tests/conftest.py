import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (run via gpurun)")


@pytest.fixture(scope="session")
def oracle():
    """ctypes handle to the CPU oracle (parity referee)."""
    import ctypes

    path = os.path.join(REPO, "oracle", "liblp_oracle.so")
    assert os.path.exists(path), "build the oracle first (oracle/Makefile)"
    lib = ctypes.CDLL(path)
    lib.kmp_oracle_lp_refine.restype = ctypes.c_int64
    lib.kmp_oracle_lp_cluster.restype = ctypes.c_int64
    lib.kmp_oracle_edge_cut.restype = ctypes.c_int64
    return lib


@pytest.fixture(scope="session")
def ref():
    """ctypes handle to the compiled reference (oracle/_ref); None on boxes
    where it was not built (it is built in the dev container and ships with
    the snapshot)."""
    import ctypes

    path = os.path.join(REPO, "oracle", "_ref", "libkaminpar_ref.so")
    if not os.path.exists(path):
        return None
    lib = ctypes.CDLL(path)
    lib.kref_lp_refine.restype = ctypes.c_int64
    lib.kref_lp_cluster.restype = ctypes.c_int64
    lib.kref_edge_cut.restype = ctypes.c_int64
    lib.kref_max_block_weight.restype = ctypes.c_int64
    return lib
