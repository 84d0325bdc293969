"""Degree-bucket rearrangement (the reference's default preprocessing,
NodeOrdering::DEGREE_BUCKETS, graphutils/permutator.h:30-128): permutation
validity, bucket ordering, row preservation, cut invariance."""

import numpy as np
import pytest

import kaminpar_amd as ka


def _bucket(deg):
    return 33 - 1 if deg == 0 else int(np.floor(np.log2(deg))) + 1


@pytest.mark.parametrize("scale,seed", [(10, 7), (12, 42)])
def test_degree_bucket_permutation(scale, seed):
    g = ka.Graph.rmat(scale, 8, seed=seed)
    gp, perm = g.rearrange_degree_buckets()
    n = g.n
    assert gp.n == n and gp.m == g.m

    # bijection
    assert np.array_equal(np.sort(perm), np.arange(n, dtype=np.uint32))

    deg_old = np.diff(np.asarray(g.xadj)).astype(np.int64)
    deg_new = np.diff(np.asarray(gp.xadj)).astype(np.int64)
    # degrees carried over
    assert np.array_equal(deg_new[perm], deg_old)

    # bucket ids are non-decreasing over new vertex order (deg-0 last)
    buckets_new = np.where(deg_new == 0, 32,
                           np.floor(np.log2(np.maximum(deg_new, 1))).astype(np.int64) + 1)
    assert (np.diff(buckets_new) >= 0).all()

    # stable within bucket: for equal buckets, perm preserves old id order
    bucket_old = np.where(deg_old == 0, 32,
                          np.floor(np.log2(np.maximum(deg_old, 1))).astype(np.int64) + 1)
    for b in np.unique(bucket_old):
        members = np.flatnonzero(bucket_old == b)
        assert (np.diff(perm[members].astype(np.int64)) > 0).all()

    # adjacency rows preserved (remapped targets, same order)
    xo, ao = np.asarray(g.xadj), np.asarray(g.adjncy)
    xn, an = np.asarray(gp.xadj), np.asarray(gp.adjncy)
    rng = np.random.default_rng(0)
    for u in rng.integers(0, n, size=50):
        v = int(perm[u])
        row_old = perm[ao[xo[u]:xo[u + 1]]]
        row_new = an[xn[v]:xn[v + 1]]
        assert np.array_equal(row_old, row_new)


def test_degree_bucket_cut_invariance():
    g = ka.Graph.rmat(12, 8, seed=3)
    gp, perm = g.rearrange_degree_buckets()
    labels_old = ka.random_partition(g.n, 16, seed=9)
    labels_new = np.zeros(g.n, dtype=np.uint32)
    labels_new[perm] = labels_old  # l_new[perm[u]] = l_old[u]
    assert g.edge_cut(labels_old) == gp.edge_cut(labels_new)
    # and mapping back as documented: l_old = l_new[perm]
    assert np.array_equal(labels_new[perm], labels_old)


def test_rgg2d_generator_properties():
    """Config-4 generator: expected density, symmetric, no self loops."""
    g = ka.Graph.rgg2d(1 << 14, avg_deg=16.0, seed=42)
    assert g.n == 1 << 14
    avg = g.m / g.n
    assert 10.0 < avg < 24.0, avg  # m counts directed arcs; avg_deg ~16
    xadj = np.asarray(g.xadj)
    adjncy = np.asarray(g.adjncy)
    u = np.repeat(np.arange(g.n, dtype=np.uint64), np.diff(xadj))
    assert not (u == adjncy).any()  # no self loops
    fwd = set(zip(u.tolist(), adjncy.tolist()))
    assert all((v, w) in fwd for (w, v) in list(fwd)[:2000])  # symmetric
