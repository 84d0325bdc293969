"""ParHIP binary format round trip (docs/graph_file_format.md), including
weighted graphs and 64-bit-stored-id files that fit u32."""

import os
import struct

import numpy as np

import kaminpar_amd as ka


def test_parhip_roundtrip_unweighted(tmp_path):
    g = ka.Graph.rmat(12, 8, 42)
    p = str(tmp_path / "g.parhip")
    g.write_parhip(p)
    h = ka.Graph.read_parhip(p)
    assert h.n == g.n and h.m == g.m
    assert np.array_equal(np.asarray(h.xadj), np.asarray(g.xadj))
    assert np.array_equal(np.asarray(h.adjncy), np.asarray(g.adjncy))


def test_parhip_roundtrip_weighted(tmp_path):
    rng = np.random.default_rng(7)
    g0 = ka.Graph.rmat(10, 8, 1)
    vw = rng.integers(1, 50, g0.n).astype(np.int32)
    aw = np.ones(g0.m, np.int32)
    g = ka.Graph.from_csr(np.asarray(g0.xadj), np.asarray(g0.adjncy), vwgt=vw, adjwgt=aw)
    p = str(tmp_path / "gw.parhip")
    g.write_parhip(p)
    h = ka.Graph.read_parhip(p)
    assert h.total_node_weight == int(vw.sum())
    from kaminpar_amd import _lib
    hv = np.ctypeslib.as_array(_lib.kmp_graph_vwgt(h._h), shape=(h.n,))
    assert np.array_equal(hv, vw)
    labels = ka.random_partition(g.n, 4, seed=3)
    assert g.edge_cut(labels) == h.edge_cut(labels)


def test_parhip_reads_64bit_stored_ids(tmp_path):
    """A file written with 64-bit offsets/ids (version bits cleared) loads
    into the u32 container when the values fit."""
    n, arcs = 3, [(0, 1), (1, 0), (1, 2), (2, 1)]
    xadj = [0, 1, 3, 4]
    adjncy = [v for _, v in arcs]
    p = str(tmp_path / "wide.parhip")
    with open(p, "wb") as f:
        version = 1 | 2  # no weights; 64-bit edge ids + node ids (bits 2,3 = 0)
        f.write(struct.pack("<QQQ", version, n, len(adjncy)))
        base = 24 + (n + 1) * 8
        for o in xadj:
            f.write(struct.pack("<Q", base + o * 8))
        for v in adjncy:
            f.write(struct.pack("<Q", v))
    h = ka.Graph.read_parhip(p)
    assert h.n == n and h.m == 4
    assert list(h.xadj) == xadj
    assert list(h.adjncy) == adjncy


def test_c_shim_fails_loudly_without_gpu():
    """kaminpar_amd_compute_partition returns -1 (no silent CPU fallback)
    when no HIP device is available; skipped if a GPU is present."""
    import ctypes
    import torch
    if torch.cuda.is_available():
        import pytest
        pytest.skip("GPU present")
    from kaminpar_amd import _lib, _u32p
    lib = ctypes.CDLL(None)  # symbols are in the already-loaded library
    h = _lib_shim = _lib  # reuse loaded lib handle for the shim symbols
    _lib.kaminpar_amd_create.restype = ctypes.c_void_p
    _lib.kaminpar_amd_compute_partition.restype = ctypes.c_int64
    _lib.kaminpar_amd_compute_partition.argtypes = [ctypes.c_void_p,
                                                    ctypes.POINTER(ctypes.c_uint32)]
    shm = _lib.kaminpar_amd_create(1)
    g = ka.Graph.rmat(10, 8, 1)
    _lib.kaminpar_amd_copy_graph.argtypes = [
        ctypes.c_void_p, ctypes.c_uint32, ctypes.POINTER(ctypes.c_uint32),
        ctypes.POINTER(ctypes.c_uint32), ctypes.c_void_p, ctypes.c_void_p]
    xadj = np.ascontiguousarray(g.xadj)
    adjncy = np.ascontiguousarray(g.adjncy)
    _lib.kaminpar_amd_copy_graph(shm, g.n, _u32p(xadj), _u32p(adjncy), None, None)
    _lib.kaminpar_amd_set_k.argtypes = [ctypes.c_void_p, ctypes.c_uint32]
    _lib.kaminpar_amd_set_k(shm, 4)
    part = np.zeros(g.n, np.uint32)
    cut = _lib.kaminpar_amd_compute_partition(shm, _u32p(part))
    assert cut == -1
    _lib.kaminpar_amd_free.argtypes = [ctypes.c_void_p]
    _lib.kaminpar_amd_free(shm)


def test_metis_roundtrip(tmp_path):
    g = ka.Graph.rmat(10, 8, 3)
    p = str(tmp_path / "g.metis")
    g.write_metis(p)
    h = ka.Graph.read_metis(p)
    assert h.n == g.n and h.m == g.m
    assert np.array_equal(np.asarray(h.xadj), np.asarray(g.xadj))
    assert np.array_equal(np.asarray(h.adjncy), np.asarray(g.adjncy))


def test_metis_roundtrip_weighted(tmp_path):
    rng = np.random.default_rng(4)
    g0 = ka.Graph.rmat(9, 8, 5)
    vw = rng.integers(1, 9, g0.n).astype(np.int32)
    g = ka.Graph.from_csr(np.asarray(g0.xadj), np.asarray(g0.adjncy), vwgt=vw)
    p = str(tmp_path / "gw.metis")
    g.write_metis(p)
    h = ka.Graph.read_metis(p)
    assert h.total_node_weight == int(vw.sum())
    labels = ka.random_partition(g.n, 3, seed=1)
    assert g.edge_cut(labels) == h.edge_cut(labels)
