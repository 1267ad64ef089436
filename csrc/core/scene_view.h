// scene_view.h — the flat, pointer-based view of a scene that kernels and the
// CPU reference renderer consume.  All arrays are built by the Python Scene
// layer (hippt/scene/scene.py) as torch/numpy buffers and passed as raw
// pointers; the struct goes to the GPU by value via kernarg.
//
// Capability parity: reference src/core/scene.cuh (Scene resource ownership,
// export_prims), object.cuh (ObjInfo / CompactedObjInfo), max_depth.h
// (MaxDepthParams incl. ToF min/max time).
#pragma once
#include "bvh.h"
#include "bvh4.h"
#include "bvh8.h"
#include "bsdf.h"
#include "emitter.h"
#include "camera.h"
#include "medium.h"

namespace hippt {

struct alignas(16) ObjInfo {
    int32_t prim_base;
    int32_t prim_cnt;
    int32_t bsdf_id;
    int32_t emitter_id;   // -1 = not an emitter
    int32_t medium_in;    // interior medium id, -1 = vacuum
    int32_t medium_out;   // exterior medium id, -1 = vacuum
    uint32_t flags;       // bit0 = cullable (alpha-masked forward boundary)
    float inv_area;       // 1 / total surface area
};
constexpr uint32_t OBJ_CULLABLE = 1u;

struct MaxDepthParams {
    int max_depth;     // total bounce cap
    int max_diffuse;
    int max_specular;
    int max_transmit;
    int max_volume;
    float min_time;    // ToF gating window (SUPPORTS_TOF_RENDERING parity)
    float max_time;
    int use_tof;
    float radiance_clamp;  // per-sample radiance cap, 0 = off (firefly knob)
};

struct SceneView {
    // geometry + BVH (binary skip-link kept for the cost visualizer and as
    // the collapse source; traversal runs on the 4-wide tree when present)
    const BVHNode* nodes; int n_nodes;
    const BVH4Node* nodes4; int n_nodes4;
    const BVH4NodeQ* nodes4q;               // quantized mirror (HIPPT_QBVH)
    const BVH8Node* nodes8; int n_nodes8;   // A/B: used when non-null
    const Prim* prims; const PrimAttr* attrs; const uint32_t* prim_obj; int n_prims;
    const ObjInfo* objs; int n_objs;
    // materials / emitters / textures
    const BsdfParams* bsdfs; int n_bsdfs;
    const EmitterParams* emitters; int n_emitters;
    const int* emitter_prims;
    const float* emitter_cdf;
    // envmap luminance-CDF tables (importance-sampled NEE; null = cosine)
    const float* env_rows; const float* env_cols; int env_w, env_h;
    // power-proportional light selection CDF (null = uniform, reference)
    const float* emitter_sel_cdf;
    const TexView* textures; int n_textures;
    int env_emitter;   // emitter index of the EM_ENVMAP, or -1
    // media
    const MediumParams* media; int n_media;
    const PhaseParams* phases;
    int cam_medium;    // medium the camera sits in (-1 = vacuum)
    // camera + depth caps
    Camera cam;
    MaxDepthParams md;
    // max depth of the 4-wide tree (levels); gates kernels with halved
    // traversal stacks (wavefront dual-walk trace)
    int bvh4_depth;
    // scene bounding sphere (xyz = center, w = radius); envmap sample_le
    Vec4 scene_bound;
    // no spheres in the scene: leaf tests skip the per-prim prim_obj load
    // (reference TRIANGLE_ONLY compile flag as a scene-uniform branch)
    int tri_only;
    // LDS top-tree cache size in nodes, from the accelerator XML cache_level
    // (reference semantics: top 2^level binary nodes cached; here 4-wide
    // nodes).  0 = launcher default; HIPPT_TOPCACHE env still overrides.
    int cache_nodes;

    HD EmitterGeom emitter_geom() const {
        return {prims, attrs, prim_obj, emitter_prims, emitter_cdf, textures,
                env_rows, env_cols, env_w, env_h,
                scene_bound.xyz(), scene_bound.w};
    }
    HD uint32_t obj_of_prim(int pid) const { return prim_obj[pid] & PRIM_OBJ_MASK; }
    HD bool prim_is_sphere(int pid) const { return (prim_obj[pid] & PRIM_SPHERE_BIT) != 0; }
};

// Light selection: power-proportional when the CDF is present (extension;
// the reference picks uniformly), uniform otherwise.  Any positive weights
// keep the estimator unbiased because the pick pdf divides the sample.
HD float emitter_sel_pdf(const SceneView& sv, int i) {
    if (sv.n_emitters <= 0) return 0.f;
    if (!sv.emitter_sel_cdf) return 1.f / sv.n_emitters;
    return sv.emitter_sel_cdf[i] - (i ? sv.emitter_sel_cdf[i - 1] : 0.f);
}

HD int pick_emitter(const SceneView& sv, Sampler& sp, float& pdf) {
    if (sv.n_emitters <= 0) { pdf = 0.f; return -1; }
    if (sv.emitter_sel_cdf) {
        int i = cdf_find(sv.emitter_sel_cdf, sv.n_emitters, sp.next1f());
        pdf = emitter_sel_pdf(sv, i);
        return i;
    }
    int i = (int)(sp.next1f() * sv.n_emitters);
    i = i >= sv.n_emitters ? sv.n_emitters - 1 : i;
    pdf = 1.f / sv.n_emitters;
    return i;
}

// Traversal entry points used by every integrator/kernel.  The 4-wide
// ordered walk is the ONLY device path — keeping the binary walk as a
// runtime fallback was measured to cost the megakernel ~11% (dead code +
// its register pressure under __launch_bounds__); the binary tree stays as
// the collapse source and for host-side self-tests (tests/test_core.py).
// Per-thread traversal-stack context: device kernels point lds_slot into a
// __shared__ array (entry d at lds_slot[d*256]); host passes the default
// (pure private stack).  See bvh4.h for why the stack lives in LDS.
// Compile-time traversal-tree selection: HIPPT_QBVH makes the 64-byte
// quantized node (bvh4.h BVH4NodeQ) the tree every kernel walks — a
// runtime branch here was measured to cost ~3% even when dead (see the
// BVH8 note above), so the choice is a build flag, A/B'd by rebuilding.
// MEASURED AND REJECTED as the default (kitchen 1080p megakernel: 99.6
// vs 146.5 Msps): the walk is bound by dependent-load LATENCY, not
// bandwidth, so halving node bytes buys nothing while the uint8
// decompress (cvt+fma per bound) sits ON the critical path between the
// node load and the slab tests.  The Q tree stays host-verified
// (tests/test_core.py) as a recorded experiment.
#ifdef HIPPT_QBVH
using TravNode = BVH4NodeQ;
#else
using TravNode = BVH4Node;
#endif

struct TravCtx {
    uint64_t* lds_slot = nullptr;
    int lds_n = 0;
    const TravNode* top_cache = nullptr;  // LDS copy of the tree top
    int n_cached = 0;
};

HD const TravNode* trav_nodes(const SceneView& sv) {
#ifdef HIPPT_QBVH
    return sv.nodes4q;
#else
    return sv.nodes4;
#endif
}

// Closest hit = the while-while phase-batched walk (measured +25% megakernel
// / +19% wavefront over the inline-leaf ordered walk, profiles/README.md);
// the inline walk stays for host-side self-tests.
// BVH8 (bvh8.h) was measured 2.4x SLOWER than BVH4 here (256-byte node
// copy = 64 VGPRs of temporaries + SAH dilution at width 8), so the 8-wide
// walk is NOT wired into the kernels — even a dead `if (sv.nodes8)` branch
// cost ~3% of kernel throughput.  bvh8.h stays host-tested for the record.
// (A 4-byte-entry stack variant — double LDS capacity, no pop culling —
// was measured 3% slower: the t_near culling pays for its 8-byte entries.
// ray_intersect_bvh4_ww32 stays host-tested for the record.)
HD HitRecord scene_intersect(const SceneView& sv, const Ray& ray,
                             float tmax = MAX_DIST, TravCtx tc = {}) {
#ifdef HIPPT_QBVH
    return ray_intersect_bvh4q_ww(sv.nodes4q, sv.prims, sv.prim_obj, ray, tmax,
                                  tc.lds_slot, tc.lds_n, tc.top_cache, tc.n_cached);
#else
    return ray_intersect_bvh4_ww(sv.nodes4, sv.prims, sv.prim_obj, ray, tmax,
                                 tc.lds_slot, tc.lds_n, tc.top_cache, tc.n_cached,
                                 sv.tri_only != 0);
#endif
}
// Any-hit also runs the phase-batched form (+1-2% measured over the
// inline-leaf walk; the inline walk stays for host self-tests).
HD bool scene_occluded(const SceneView& sv, const Ray& ray, float tmax,
                       TravCtx tc = {}) {
#ifdef HIPPT_QBVH
    return occlusion_test_bvh4q_ww(sv.nodes4q, sv.prims, sv.prim_obj, ray, tmax,
                                   tc.lds_slot, tc.lds_n, tc.top_cache, tc.n_cached);
#else
    return occlusion_test_bvh4_ww(sv.nodes4, sv.prims, sv.prim_obj, ray, tmax,
                                  tc.lds_slot, tc.lds_n, tc.top_cache, tc.n_cached,
                                  sv.tri_only != 0);
#endif
}

} // namespace hippt
