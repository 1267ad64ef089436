"""Property-based tests (hypothesis): the native BVH builder + 4-wide walk
must agree with an INDEPENDENT numpy Möller-Trumbore brute force on random
triangle soups and random rays — construction invariants (leaf coverage,
reordering, collapse) and traversal correctness in one property.  The
brute force is implemented here with numpy only (no csrc math)."""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

from hippt import C


def brute_force_t(tris, o, d):
    """Vectorized Möller-Trumbore over all triangles; returns min positive t
    (or inf).  tris: (n,3,3) vertices; o,d: (3,)."""
    v0 = tris[:, 0]
    e1 = tris[:, 1] - tris[:, 0]
    e2 = tris[:, 2] - tris[:, 0]
    p = np.cross(d, e2)
    det = np.einsum("ij,ij->i", e1, p)
    ok = np.abs(det) > 1e-12
    inv = np.where(ok, 1.0 / np.where(ok, det, 1.0), 0.0)
    s = o - v0
    u = np.einsum("ij,ij->i", s, p) * inv
    q = np.cross(s, e1)
    v = np.einsum("j,ij->i", d, q) * inv
    t = np.einsum("ij,ij->i", e2, q) * inv
    hit = ok & (u >= -1e-7) & (v >= -1e-7) & (u + v <= 1 + 1e-7) & (t > 1e-3)
    return t[hit].min() if hit.any() else np.inf


def make_prims(tris):
    n = len(tris)
    prims = np.zeros((n, 12), np.float32)
    prims[:, 0:3] = tris[:, 0]
    prims[:, 4:7] = tris[:, 1] - tris[:, 0]
    prims[:, 8:11] = tris[:, 2] - tris[:, 0]
    return prims


@settings(max_examples=25, deadline=None)
@given(seed=st.integers(0, 2**31 - 1), n=st.integers(1, 160),
       clustered=st.booleans(), use_sbvh=st.booleans())
def test_bvh4_walk_matches_numpy_brute_force(seed, n, clustered, use_sbvh):
    rng = np.random.default_rng(seed)
    if clustered:
        # clustered blobs stress SAH binning + spatial splits
        centers = rng.uniform(-4, 4, (max(1, n // 16), 3))
        base = centers[rng.integers(0, len(centers), n)]
        tris = base[:, None, :] + rng.normal(0, 0.25, (n, 3, 3))
    else:
        tris = rng.uniform(-5, 5, (n, 1, 3)) + rng.normal(0, 0.7, (n, 3, 3))
    tris = tris.astype(np.float32)
    prims = make_prims(tris)
    prim_obj = np.zeros(n, np.uint32)
    nodes, order, stats = C.build_bvh(prims, prim_obj, 8, 0.6, use_sbvh, True, 1.0)
    # SBVH may duplicate references; gather through order
    rp = np.ascontiguousarray(prims[order])
    rpo = np.ascontiguousarray(prim_obj[order])
    nodes4, depth4 = C.collapse_bvh4(nodes)
    assert depth4 >= 1
    o = rng.uniform(-8, 8, (64, 3)).astype(np.float32)
    d = rng.normal(size=(64, 3)).astype(np.float32)
    d /= np.linalg.norm(d, axis=1, keepdims=True)
    t_bvh, p_bvh = C.bvh4_hit(rp, rpo, nodes4, o, d)
    for i in range(len(o)):
        t_ref = brute_force_t(tris.astype(np.float64), o[i].astype(np.float64),
                              d[i].astype(np.float64))
        if np.isinf(t_ref):
            assert p_bvh[i] < 0 or t_bvh[i] > 1e6, (i, t_bvh[i], t_ref)
        else:
            assert p_bvh[i] >= 0, (i, "BVH missed a hit at t", t_ref)
            assert abs(t_bvh[i] - t_ref) < 1e-3 * max(1.0, t_ref), \
                (i, t_bvh[i], t_ref)


@settings(max_examples=15, deadline=None)
@given(seed=st.integers(0, 2**31 - 1),
       shape=st.tuples(st.integers(1, 40), st.integers(1, 40), st.integers(1, 40)),
       vs=st.floats(0.01, 10.0))
def test_nvdb_roundtrip_property(seed, shape, vs):
    from hippt.scene.nvdb import write_nvdb, read_nvdb
    import tempfile, os
    rng = np.random.default_rng(seed)
    d = (rng.random(shape, np.float32) * (rng.random(shape) > 0.5)).astype(np.float32)
    f = tempfile.NamedTemporaryFile(suffix=".nvdb", delete=False)
    f.close()
    try:
        write_nvdb(f.name, d, voxel_size=vs,
                   origin=tuple(int(v) for v in rng.integers(0, 64, 3)),
                   world_origin=tuple(rng.uniform(-5, 5, 3)))
        g = read_nvdb(f.name)[0]
        np.testing.assert_array_equal(g["dense"], d)
    finally:
        os.unlink(f.name)
