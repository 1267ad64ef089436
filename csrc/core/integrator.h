// integrator.h — the unidirectional path-tracing core (surface scenes).
// Single-source: this exact function body is the CPU reference renderer
// (OpenMP over pixels) and the body of the GPU megakernel.
//
// Capability parity: reference src/pt_impl/megakernel_pt.cu:33-201
// (render_pt_kernel: BVH hit -> MIS emitter-hit accumulation -> NEE with one
// emitter sample + shadow ray -> BSDF sample -> per-lobe depth caps + RR with
// threshold 0.1 after bounce 1) and the ToF time gating of megakernel_vpt.cu
// (CONDITION_BLOCK time_in_range).
#pragma once
#include "scene_view.h"

namespace hippt {

struct PathStats {
    int n_diffuse = 0, n_specular = 0, n_transmit = 0, n_volume = 0;
};

HD bool tof_in_range(const MaxDepthParams& md, float time) {
    return !md.use_tof || (time >= md.min_time && time <= md.max_time);
}

// Resumable per-path state: one `path_step` = one bounce of the
// unidirectional path loop.  Shared by trace_path (per-sample form) and the
// megakernel's path-regeneration loop (k_render starts the next sample the
// moment a lane's path dies instead of idling until the wave's longest path
// finishes — wave64 tail divergence was the measured bottleneck,
// VALUUtilization ~15-20%, profiles/README.md).
struct PathState {
    Ray ray;
    Vec3 L, thp;
    float prev_pdf;
    bool prev_delta;
    Vec3 prev_n;
    float path_time;
    PathStats st;
    int b, iter;
    // primary-hit AOVs (denoiser guides): shading normal, depth, albedo
    Vec3 aov_n, aov_alb;
    float aov_t;
    float lambda;   // per-path dispersion wavelength (0 = unsampled)

    HD void reset(const Ray& r) {
        ray = r;
        L = Vec3(0.f); thp = Vec3(1.f);
        prev_pdf = 0.f;
        prev_delta = true;
        prev_n = Vec3(0.f, 0.f, 1.f);
        path_time = 0.f;
        st = PathStats();
        b = 0; iter = 0;
        aov_n = Vec3(0.f); aov_alb = Vec3(0.f); aov_t = 0.f;
        lambda = 0.f;
    }
};

// First-hit albedo proxy per BSDF family (denoiser guide, not physics).
HD Vec3 bsdf_aov_albedo(const BsdfParams& b, Vec2 uv, const TexView* textures) {
    switch (b.type) {
    case BSDF_LAMBERTIAN:
    case BSDF_PLASTIC:
    case BSDF_PLASTIC_FORWARD:
        return bsdf_albedo(b, uv, textures);
    case BSDF_GGX_CONDUCTOR:
        return tex_or(textures, b.tex[TEX_SPECULAR], uv, b.kg.xyz());
    case BSDF_SPECULAR:
    case BSDF_TRANSLUCENT:
    case BSDF_DISPERSION:
        return tex_or(textures, b.tex[TEX_SPECULAR], uv, b.ks.xyz());
    default:
        return Vec3(1.f);
    }
}

// Shade one already-intersected hit (miss handling included): emitter-hit
// MIS, NEE, BSDF sampling, caps + RR.  Split out of path_step so the
// wavefront tail kernel (wf_kernels.hip k_wf_tail) can resume a path whose
// current hit record is already in the payload pool.  Returns true when the
// path is finished (L is final).
HD bool path_shade_hit(const SceneView& sv, PathState& ps, Sampler& sp, TravCtx tc,
                       const HitRecord& hit) {
    Ray& ray = ps.ray;
    Vec3& L = ps.L;
    Vec3& thp = ps.thp;
    float& prev_pdf = ps.prev_pdf;
    bool& prev_delta = ps.prev_delta;
    Vec3& prev_n = ps.prev_n;
    float& path_time = ps.path_time;
    PathStats& st = ps.st;
    int& b = ps.b;
    {
        if (hit.prim_idx < 0) {
            // miss -> environment map with MIS against the cosine NEE pdf
            if (sv.env_emitter >= 0) {
                const EmitterParams& env = sv.emitters[sv.env_emitter];
                Vec3 le = envmap_eval(env, ray.d, sv.textures);
                float w = 1.f;
                if (!prev_delta) {
                    float light_pdf = emitter_pdf_hit(env, ray.d, ENVMAP_DIST, ray.d, prev_n, sv.emitter_geom()) *
                                      emitter_sel_pdf(sv, sv.env_emitter);
                    w = mis_weight(prev_pdf, light_pdf);
                }
                if (tof_in_range(sv.md, path_time + ENVMAP_DIST)) L += thp * le * w;
            }
            return true;
        }
        Vec3 pos = ray.at(hit.t);
        path_time += hit.t;
        uint32_t po = sv.prim_obj[hit.prim_idx];
        bool is_sphere = (po & PRIM_SPHERE_BIT) != 0;
        const ObjInfo& obj = sv.objs[po & PRIM_OBJ_MASK];
        const Prim prim = sv.prims[hit.prim_idx];
        Interaction it = get_interaction(prim, sv.attrs[hit.prim_idx], is_sphere, pos, hit.u, hit.v);
        const BsdfParams& bsdf = sv.bsdfs[obj.bsdf_id];
        if (bsdf.tex[TEX_NORMAL] >= 0)
            it.shading_n = apply_normal_map(sv.textures, bsdf.tex[TEX_NORMAL], it.uv, it.shading_n);
        if (ps.iter == 1) {
            ps.aov_n = it.shading_n;
            ps.aov_t = hit.t;
            ps.aov_alb = bsdf_aov_albedo(bsdf, it.uv, sv.textures);
        }

        // ---- emitter hit accumulation with MIS (megakernel_pt.cu:91-152)
        if (obj.emitter_id >= 0) {
            const EmitterParams& em = sv.emitters[obj.emitter_id];
            Vec3 le = emitter_eval_le(em, it.shading_n, -ray.d, it.uv, sv.textures);
            if (!le.is_zero()) {
                float w = 1.f;
                if (!prev_delta) {
                    float light_pdf = emitter_pdf_hit(em, ray.d, hit.t, it.shading_n, prev_n, sv.emitter_geom()) *
                                   emitter_sel_pdf(sv, obj.emitter_id);
                    w = mis_weight(prev_pdf, light_pdf);
                }
                if (tof_in_range(sv.md, path_time)) L += thp * le * w;
            }
        }

        // ---- next-event estimation (one emitter sample + shadow ray)
        if (!bsdf_is_delta(bsdf) && sv.n_emitters > 0) {
            float epdf;
            int ei = pick_emitter(sv, sp, epdf);
            EmitterSampleRec er = emitter_sample(sv.emitters[ei], sv.emitter_geom(), pos, it.shading_n, sp);
            if (er.pdf > 0.f && !er.radiance.is_zero()) {
                Vec3 to_l = er.pos - pos;
                float dist = to_l.length();
                Vec3 wi = to_l * (1.f / fmaxf(dist, 1e-9f));
                Vec3 f = bsdf_eval(bsdf, -ray.d, wi, it, sv.textures);
                if (!f.is_zero()) {
                    Ray sh_ray(fmadd(wi, EPSILON, pos), wi);
                    float sh_max = (sv.emitters[ei].type == EM_ENVMAP ? ENVMAP_DIST : dist) - 2.f * EPSILON;
                    if (!scene_occluded(sv, sh_ray, sh_max, tc)) {
                        float light_pdf = er.pdf * epdf;
                        float w = er.delta ? 1.f
                                           : mis_weight(light_pdf, bsdf_pdf(bsdf, -ray.d, wi, it, sv.textures));
                        if (tof_in_range(sv.md, path_time + dist))
                            L += thp * f * er.radiance * (w / light_pdf);
                    }
                }
            }
        }

        // ---- BSDF sampling
        BsdfSample bs = bsdf_sample(bsdf, -ray.d, it, sp, sv.textures, &ps.lambda);
        if (bs.pdf <= 0.f || bs.weight.is_zero()) return true;
        if (bs.weight.has_nan() || bs.wi.has_nan()) return true;  // numeric scrub
        thp *= bs.weight;

        // per-lobe bounce caps (reference max_depth.h semantics)
        if (!(bs.lobe & LOBE_NULL)) {
            if (bs.lobe & LOBE_DIFFUSE)  { if (++st.n_diffuse  > sv.md.max_diffuse)  return true; }
            if (bs.lobe & LOBE_SPECULAR) { if (++st.n_specular > sv.md.max_specular) return true; }
            if (bs.lobe & LOBE_TRANSMIT) { if (++st.n_transmit > sv.md.max_transmit) return true; }
            ++b;
        }
        prev_delta = (bs.lobe & LOBE_DELTA) != 0;
        prev_pdf = bs.pdf;
        prev_n = it.shading_n;
        ray = Ray(fmadd(bs.wi, EPSILON, pos), bs.wi);

        // Russian roulette after bounce 1, threshold 0.1 (megakernel_pt.cu RR)
        if (b > 1) {
            float p = clampv(thp.max_elem(), 0.f, 1.f);
            if (p < 0.1f) {
                if (sp.next1f() >= p * 10.f) return true;
                thp *= (1.f / (p * 10.f));
            }
        }
    }
    return false;
}

// One bounce; returns true when the path is finished (L is final).
HD bool path_step(const SceneView& sv, PathState& ps, Sampler& sp, TravCtx tc) {
    if (ps.iter >= sv.md.max_depth * 2 + 8 || ps.b >= sv.md.max_depth) return true;
    ++ps.iter;
    HitRecord hit = scene_intersect(sv, ps.ray, MAX_DIST, tc);
    return path_shade_hit(sv, ps, sp, tc, hit);
}

// Full path trace for one camera ray. Returns radiance estimate.
// Per-sample radiance cap (md.radiance_clamp > 0): the standard biased
// firefly control — applied at sample commit, never inside the estimator.
HD Vec3 clamp_radiance(const SceneView& sv, Vec3 L) {
    float c = sv.md.radiance_clamp;
    if (c > 0.f) L = L.minv(Vec3(c));
    return L;
}

HD Vec3 trace_path(const SceneView& sv, Ray ray, Sampler& sp, TravCtx tc = {}) {
    PathState ps;
    ps.reset(ray);
    while (!path_step(sv, ps, sp, tc)) {}
    if (ps.L.has_nan()) return Vec3(0.f);
    return clamp_radiance(sv, ps.L);
}

// Depth renderer: distance of the primary hit (reference pt_impl/depth.cu).
HD float trace_depth(const SceneView& sv, const Ray& ray, TravCtx tc = {}) {
    HitRecord hit = scene_intersect(sv, ray, MAX_DIST, tc);
    return hit.prim_idx >= 0 ? hit.t : 0.f;
}

// BVH-cost visualizer: counts node visits (x) and primitive tests (y) for a
// primary ray (reference pt_impl/bvh_cost.cu:38-101).
HD Vec2 trace_bvh_cost(const SceneView& sv, const Ray& ray) {
    return bvh4_cost(sv.nodes4, sv.prims, sv.prim_obj, ray);
}
HD Vec2 trace_bvh_cost_binary(const SceneView& sv, const Ray& ray) {
    Vec3 inv_d = safe_rcp_dir(ray.d);
    Vec3 o_div = ray.o * inv_d;
    float best_t = MAX_DIST;
    int node_visits = 0, prim_tests = 0;
    int i = 0;
    while (i < sv.n_nodes) {
        const BVHNode nd = sv.nodes[i];
        ++node_visits;
        float t_near;
        bool hit_box = nd.aabb().intersect(inv_d, o_div, best_t, t_near);
        int cnt = float_as_int(nd.hi.w);
        if (hit_box) {
            if (cnt > 0) {
                int base = nd.prim_base();
                for (int k = 0; k < cnt; ++k) {
                    ++prim_tests;
                    float u, v;
                    bool sph = (sv.prim_obj[base + k] & PRIM_SPHERE_BIT) != 0;
                    float t = intersect_prim(sv.prims[base + k], sph, ray, u, v);
                    if (t > EPSILON && t < best_t) best_t = t;
                }
            }
            ++i;
        } else {
            i = cnt > 0 ? i + 1 : -cnt;
        }
    }
    return {(float)node_visits, (float)prim_tests};
}

} // namespace hippt
