"""Core invariants: native struct layouts, RNG determinism, BVH build/traverse."""
import numpy as np
import pytest

import hippt
from hippt import C


def test_struct_sizes():
    s = dict(C.struct_sizes())
    assert s["BVHNode"] == 32
    assert s["Prim"] == 48
    assert s["PrimAttr"] == 64
    assert s["ObjInfo"] == 32
    assert s["BsdfParams"] == 80
    assert s["EmitterParams"] == 64
    assert s["PhaseParams"] == 16
    assert s["Camera"] % 16 == 0
    assert C.WAVE_SIZE == 64


def _random_tris(n, seed=0, spread=10.0):
    rng = np.random.default_rng(seed)
    centers = rng.uniform(-spread, spread, (n, 1, 3))
    tris = centers + rng.uniform(-0.5, 0.5, (n, 3, 3))
    return tris.astype(np.float32)


def _to_prims(tris):
    n = len(tris)
    pr = np.zeros((n, 12), np.float32)
    pr[:, 0:3] = tris[:, 0]
    pr[:, 4:7] = tris[:, 1] - tris[:, 0]
    pr[:, 8:11] = tris[:, 2] - tris[:, 0]
    return pr


class TestBVH:
    def test_leaves_partition_prims(self):
        tris = _random_tris(500, seed=1)
        prims = _to_prims(tris)
        pobj = np.zeros(500, np.uint32)
        nodes, order, stats = C.build_bvh(prims, pobj, 4, 0.0, False, True)
        # prim_order is a permutation
        assert sorted(order.tolist()) == list(range(500))
        # leaf ranges cover [0, n) exactly once
        lo_w = nodes[:, 3].view(np.int32)
        hi_w = nodes[:, 7].view(np.int32)
        covered = np.zeros(500, bool)
        for i in range(len(nodes)):
            if hi_w[i] > 0:  # leaf
                b, c = lo_w[i], hi_w[i]
                assert not covered[b:b + c].any()
                covered[b:b + c] = True
        assert covered.all()

    def test_skip_links_valid(self):
        tris = _random_tris(300, seed=2)
        nodes, order, stats = C.build_bvh(_to_prims(tris), np.zeros(300, np.uint32),
                                          4, 0.0, False, True)
        n = len(nodes)
        hi_w = nodes[:, 7].view(np.int32)
        for i in range(n):
            if hi_w[i] <= 0:  # internal: skip strictly after i, within bounds
                skip = -hi_w[i]
                assert i + 2 < skip <= n  # at least two children inside

    def test_child_boxes_inside_parent(self):
        tris = _random_tris(200, seed=3)
        nodes, order, stats = C.build_bvh(_to_prims(tris), np.zeros(200, np.uint32),
                                          4, 0.0, False, True)
        hi_w = nodes[:, 7].view(np.int32)
        # root box contains all prims
        lo_root, hi_root = nodes[0, 0:3], nodes[0, 4:7]
        assert (tris.reshape(-1, 3) >= lo_root - 1e-4).all()
        assert (tris.reshape(-1, 3) <= hi_root + 1e-4).all()

    def test_traversal_matches_bruteforce(self):
        """Depth through BVH equals brute-force min-t over all triangles."""
        tris = _random_tris(300, seed=4, spread=3.0)
        from hippt.scene.scene import Scene, SceneDesc, ObjectDesc, BsdfDesc, CameraDesc, RenderConfig
        d = SceneDesc()
        d.bsdfs = [BsdfDesc()]
        d.objects = [ObjectDesc(tris=tris, bsdf=0)]
        d.camera = CameraDesc(pos=(0, 0, -12), lookat=(0, 0, 0), fov=40,
                              width=64, height=64)
        d.config = RenderConfig(renderer="depth", spp=1, max_depth=1)
        r = hippt.PythonRenderer(d, device_id=-1)
        depth = r.render(spp=1).numpy()[:, :, 0]

        # brute force: Moller-Trumbore per pixel center ray... use many rays
        # through the same camera model is hard to replicate exactly w/ jitter,
        # so check a statistical property instead: every finite depth must be
        # achievable by some triangle (within scene bounds along the ray).
        finite = depth[depth > 0]
        assert finite.size > 0
        assert finite.min() > 8.0  # camera sits 12 away from a 4-unit cloud
        assert finite.max() < 16.0


class TestRNG:
    def test_deterministic(self):
        from hippt.scene.procedural import cornell_box
        d = cornell_box(width=32, height=32, spp=2, max_depth=3)
        a = hippt.PythonRenderer(d, device_id=-1).render(spp=4).numpy()
        d2 = cornell_box(width=32, height=32, spp=2, max_depth=3)
        b = hippt.PythonRenderer(d2, device_id=-1).render(spp=4).numpy()
        np.testing.assert_array_equal(a, b)

    def test_seed_offset_decorrelates(self):
        from hippt.scene.procedural import cornell_box
        d = cornell_box(width=32, height=32, spp=2, max_depth=3)
        a = hippt.PythonRenderer(d, device_id=-1, seed_offset=0).render(spp=4).numpy()
        d2 = cornell_box(width=32, height=32, spp=2, max_depth=3)
        b = hippt.PythonRenderer(d2, device_id=-1, seed_offset=1).render(spp=4).numpy()
        assert not np.allclose(a, b)
        # but the means must agree (same scene)
        assert abs(a[..., :3].mean() - b[..., :3].mean()) < 0.05 * max(a[..., :3].mean(), 1e-9)


class TestBVH4:
    """4-wide collapsed BVH (csrc/core/bvh4.h) vs binary skip-link walk."""

    def _build(self, n, seed, use_sbvh=False, spheres=False):
        tris = _random_tris(n, seed=seed, spread=3.0)
        prims = _to_prims(tris)
        pobj = np.zeros(n, np.uint32)
        if spheres:
            # last quarter become spheres: v0 = center, v0.w = radius
            k = n // 4
            prims[-k:, :] = 0
            rng = np.random.default_rng(seed + 7)
            prims[-k:, 0:3] = rng.uniform(-3, 3, (k, 3)).astype(np.float32)
            prims[-k:, 3] = rng.uniform(0.05, 0.4, k).astype(np.float32)
            pobj[-k:] = 1 << 31
        nodes, order, stats = C.build_bvh(prims, pobj, 4, 0.0, use_sbvh, True)
        prims = np.ascontiguousarray(prims[order % n])
        pobj = np.ascontiguousarray(pobj[order % n])
        nodes4, depth4 = C.collapse_bvh4(nodes)
        return prims, pobj, nodes, nodes4, depth4

    def test_collapse_invariants(self):
        prims, pobj, nodes, nodes4, depth4 = self._build(800, seed=11)
        meta = nodes4.reshape(-1, 32)
        child = meta[:, 24:28].view(np.int32)
        cnt = meta[:, 28:32].view(np.int32)
        # internal children point forward (DFS order), inside the array
        internal = child > 0
        assert (child[internal] < len(nodes4)).all()
        # leaf prim ranges cover [0, n) exactly once
        covered = np.zeros(len(prims), bool)
        for i in range(len(nodes4)):
            for c in range(4):
                if child[i, c] < 0 and cnt[i, c] > 0:
                    b, k = ~child[i, c], cnt[i, c]
                    assert not covered[b:b + k].any()
                    covered[b:b + k] = True
        assert covered.all()
        # stack bound honored
        assert 3 * depth4 <= 64

    def _rays(self, m, seed):
        rng = np.random.default_rng(seed)
        o = rng.uniform(-6, 6, (m, 3)).astype(np.float32)
        d = rng.normal(size=(m, 3)).astype(np.float32)
        d /= np.linalg.norm(d, axis=1, keepdims=True)
        # include axis-aligned rays (zero direction components: the NaN-safe
        # reciprocal path) and rays from far outside
        d[:16] = np.eye(3, dtype=np.float32)[rng.integers(0, 3, 16)] * \
            rng.choice([-1.0, 1.0], 16)[:, None]
        return o, d

    def test_traversal_agreement_bvh(self):
        prims, pobj, nodes, nodes4, _ = self._build(800, seed=12)
        o, d = self._rays(4000, seed=13)
        assert C.bvh4_selftest(prims, pobj, nodes, nodes4, o, d, 1e7) == 0

    def test_traversal_agreement_sbvh(self):
        prims, pobj, nodes, nodes4, _ = self._build(600, seed=14, use_sbvh=True)
        o, d = self._rays(4000, seed=15)
        assert C.bvh4_selftest(prims, pobj, nodes, nodes4, o, d, 1e7) == 0

    def test_traversal_agreement_spheres(self):
        prims, pobj, nodes, nodes4, _ = self._build(400, seed=16, spheres=True)
        o, d = self._rays(4000, seed=17)
        assert C.bvh4_selftest(prims, pobj, nodes, nodes4, o, d, 1e7) == 0

    def test_tiny_scene(self):
        # single-leaf degenerate tree
        prims, pobj, nodes, nodes4, depth4 = self._build(2, seed=18)
        o, d = self._rays(500, seed=19)
        assert C.bvh4_selftest(prims, pobj, nodes, nodes4, o, d, 1e7) == 0

    def test_bvh8_agreement(self):
        prims, pobj, nodes, nodes4, _ = self._build(800, seed=21)
        nodes8, depth8 = C.collapse_bvh8(nodes)
        assert 7 * depth8 <= 64
        o, d = self._rays(4000, seed=22)
        assert C.bvh4_selftest(prims, pobj, nodes, nodes4, o, d, 1e7, nodes8) == 0

    def test_bvh8_agreement_spheres(self):
        prims, pobj, nodes, nodes4, _ = self._build(400, seed=23, spheres=True)
        nodes8, depth8 = C.collapse_bvh8(nodes)
        o, d = self._rays(4000, seed=24)
        assert C.bvh4_selftest(prims, pobj, nodes, nodes4, o, d, 1e7, nodes8) == 0


class TestTraversalGroundTruth:
    """BVH4 walk vs O(n) brute force — catches tree-build errors that
    walk-vs-walk agreement cannot (both walks share the tree)."""

    def test_bvh4_vs_bruteforce(self):
        rng = np.random.default_rng(31)
        tris = _random_tris(400, seed=31, spread=1.0)
        prims = _to_prims(tris)
        pobj = np.zeros(len(tris), np.uint32)
        nodes, order, _ = C.build_bvh(prims, pobj, 4, 0.6, False, True)
        prims_r = np.ascontiguousarray(prims[order])
        pobj_r = np.ascontiguousarray(pobj[order])
        nodes4, _ = C.collapse_bvh4(nodes)
        tris_r = tris[order].astype(np.float64)

        m = 400
        o = rng.uniform(-3, 3, (m, 3))
        aim = rng.uniform(-0.8, 0.8, (m, 3))   # aim at the cloud
        d = aim - o
        d /= np.linalg.norm(d, axis=1, keepdims=True)

        # vectorized Moller-Trumbore ground truth (float64)
        v0 = tris_r[:, 0][None]                     # (1,n,3)
        e1 = (tris_r[:, 1] - tris_r[:, 0])[None]
        e2 = (tris_r[:, 2] - tris_r[:, 0])[None]
        D = d[:, None]                              # (m,1,3)
        O = o[:, None]
        p = np.cross(D, e2)
        det = np.einsum("mnk,mnk->mn", np.broadcast_to(e1, p.shape), p)
        with np.errstate(divide="ignore", invalid="ignore"):
            inv = np.where(np.abs(det) > 1e-12, 1.0 / det, 0.0)
            t0 = O - v0
            u = np.einsum("mnk,mnk->mn", t0, p) * inv
            q = np.cross(t0, np.broadcast_to(e1, t0.shape))
            v = np.einsum("mnk,mnk->mn", np.broadcast_to(D, q.shape), q) * inv
            t = np.einsum("mnk,mnk->mn", np.broadcast_to(e2, q.shape), q) * inv
        valid = (np.abs(det) > 1e-12) & (u >= 0) & (v >= 0) & (u + v <= 1) & (t > 1e-3)
        t = np.where(valid, t, np.inf)
        gt_t = t.min(axis=1)

        bt, bp = C.bvh4_hit(prims_r, pobj_r, nodes4,
                            o.astype(np.float32), d.astype(np.float32))
        hits = np.isfinite(gt_t)
        assert hits.sum() > 100
        # hit/miss agreement (tolerate fp32-vs-fp64 edge grazing)
        agree = (bp >= 0) == hits
        assert agree.mean() > 0.99, f"hit/miss disagreements: {(~agree).sum()}"
        both = hits & (bp >= 0)
        np.testing.assert_allclose(bt[both], gt_t[both], rtol=2e-3, atol=1e-3)


try:
    from hypothesis import given, settings, strategies as st

    class TestPropertyBased:
        """Property-based traversal checks (hypothesis)."""

        @given(seed=st.integers(0, 10_000), n=st.integers(2, 300),
               use_sbvh=st.booleans())
        @settings(max_examples=30, deadline=None)
        def test_any_tree_agrees(self, seed, n, use_sbvh):
            tris = _random_tris(n, seed=seed, spread=2.0)
            prims = _to_prims(tris)
            pobj = np.zeros(n, np.uint32)
            nodes, order, _ = C.build_bvh(prims, pobj, 4, 0.6, use_sbvh, True)
            pr = np.ascontiguousarray(prims[order % n])
            po = np.ascontiguousarray(pobj[order % n])
            nodes4, depth4 = C.collapse_bvh4(nodes)
            assert 3 * depth4 <= 64
            rng = np.random.default_rng(seed + 1)
            o = rng.uniform(-4, 4, (200, 3)).astype(np.float32)
            d = rng.normal(size=(200, 3)).astype(np.float32)
            d /= np.maximum(np.linalg.norm(d, axis=1, keepdims=True), 1e-12)
            assert C.bvh4_selftest(pr, po, nodes, nodes4, o, d, 1e7) == 0

        @given(lo=st.floats(-100, 100), ext=st.floats(1e-4, 100),
               seed=st.integers(0, 1000))
        @settings(max_examples=20, deadline=None)
        def test_translated_scaled_scene(self, lo, ext, seed):
            # traversal robust to arbitrary world scales/offsets
            tris = _random_tris(64, seed=seed, spread=1.0) * ext + lo
            prims = _to_prims(tris.astype(np.float32))
            pobj = np.zeros(64, np.uint32)
            nodes, order, _ = C.build_bvh(prims, pobj, 4, 0.0, False, True)
            pr = np.ascontiguousarray(prims[order])
            po = np.ascontiguousarray(pobj[order])
            nodes4, _ = C.collapse_bvh4(nodes)
            rng = np.random.default_rng(seed)
            o = (rng.uniform(-2, 2, (100, 3)) * ext + lo).astype(np.float32)
            d = rng.normal(size=(100, 3)).astype(np.float32)
            d /= np.maximum(np.linalg.norm(d, axis=1, keepdims=True), 1e-12)
            assert C.bvh4_selftest(pr, po, nodes, nodes4, o, d, 1e7) == 0
except ImportError:  # pragma: no cover
    pass


def test_bvh4_quantized_walk_matches_fp32():
    """The 64-byte quantized node (BVH4NodeQ, HIPPT_QBVH traversal tree)
    must agree with the fp32 walk on hit/miss, t, and occlusion for random
    rays over the kitchen geometry (quantized bounds round outward, so the
    closest hit is identical)."""
    from hippt.scene.procedural import kitchen
    from hippt.scene.scene import Scene
    d = kitchen(width=32, height=32)
    sc = Scene(d)
    rng = np.random.default_rng(3)
    n = 4000
    o = rng.uniform(-3, 3, (n, 3)).astype(np.float32)
    dirs = rng.normal(size=(n, 3)).astype(np.float32)
    dirs /= np.linalg.norm(dirs, axis=1, keepdims=True)
    bad = C.bvh4q_selftest(sc._np["prims"], sc._np["prim_obj"], sc._np["nodes4"],
                           o, dirs, 1e7)
    assert bad == 0, f"{bad}/{n} rays disagree"


def test_sbvh_degenerate_leaves_capped_at_15():
    """Advisor r01 (medium): SBVH fallbacks (depth cap, degenerate split)
    used to emit leaves with arbitrary prim counts, silently wrapping mod 16
    in the 4-bit traversal packing.  100 coincident triangles force every
    degenerate path; all leaves must now stay <= 15 and traversal must agree
    with brute force."""
    n = 100
    tri = np.array([[0, 0, 0], [1, 0, 0], [0, 1, 0]], np.float32)
    prims = np.zeros((n, 12), np.float32)
    prims[:, 0:3] = tri[0]
    prims[:, 4:7] = tri[1] - tri[0]
    prims[:, 8:11] = tri[2] - tri[0]
    prim_obj = np.zeros(n, np.uint32)
    nodes, order, stats = C.build_bvh(prims, prim_obj, 8, 0.6, True, True, 1.0)
    # binary leaves carry prim_cnt in hi.w as INT BITS (positive = leaf)
    cnts = nodes[:, 7].copy().view(np.int32)
    leaf_cnts = cnts[cnts > 0]
    assert len(leaf_cnts) > 0 and leaf_cnts.max() <= 15, \
        f"oversized SBVH leaf: {leaf_cnts.max() if len(leaf_cnts) else 0} prims"
    nodes4, depth4 = C.collapse_bvh4(nodes)
    rp = np.ascontiguousarray(prims[order])
    rpo = np.ascontiguousarray(prim_obj[order])
    o = np.array([[0.2, 0.2, -1.0], [0.2, 0.2, 1.0], [5, 5, 5]], np.float32)
    d = np.array([[0, 0, 1], [0, 0, -1], [0, 0, 1]], np.float32)
    t, p = C.bvh4_hit(rp, rpo, nodes4, o, d)
    assert p[0] >= 0 and abs(t[0] - 1.0) < 1e-5
    assert p[1] >= 0 and abs(t[1] - 1.0) < 1e-5
    assert p[2] < 0
