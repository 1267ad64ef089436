// light_tracer.h — adjoint (light) tracing with camera splats.
//
// Capability parity: reference src/pt_impl/megakernel_lt.cu (render_lt_kernel:
// emitter sample_le -> bounce loop -> splat to camera via get_splat_pixel with
// atomic adds; spec_constraint records only paths with >= k specular vertices
// for caustics; caustic_scaling brightness factor) + light_tracer.cu
// (alpha-count normalization, bidirectional mode drives a PT pass + LT pass —
// composed at the renderer level in hippt/render/renderer.py).
//
// The splat accumulator uses pixel-unit pinhole importance
// W = focal_px^2 / cos^3(theta_cam); contributions are pre-divided by the
// pixel count so that dividing the accumulator by its sample count (alpha
// channel) yields radiance directly comparable to the PT estimate.
#pragma once
#include "scene_view.h"
#include "integrator.h"

namespace hippt {

// SPLAT_FN: void(int pix, Vec3 value) — device uses atomics, host plain adds.
template <typename SPLAT_FN>
HD void trace_light_path_impl(const SceneView& sv, Sampler& sp, SPLAT_FN&& splat, TravCtx tc,
                              int spec_constraint, float caustic_scaling) {
    if (sv.n_emitters <= 0) return;
    float epdf;
    int ei = pick_emitter(sv, sp, epdf);
    const EmitterParams& em = sv.emitters[ei];
    EmitterLeRec le = emitter_sample_le(em, sv.emitter_geom(), sp);
    if (!le.valid) return;

    const float inv_npix = 1.f / (float(sv.cam.w) * float(sv.cam.h));
    float path_lambda = 0.f;
    const Vec3 cam_fwd = sv.cam.R * Vec3(0.f, 0.f, 1.f);

    // connect a vertex (pos, normal, f_to_cam callback result) to the camera
    auto connect = [&](const Vec3& pos, const Vec3& f_times_cos, int n_spec) {
        if (f_times_cos.is_zero()) return;
        if (n_spec < spec_constraint) return;
        int px, py;
        if (!sv.cam.get_splat_pixel(pos, px, py)) return;
        Vec3 to_cam = sv.cam.pos - pos;
        float d2 = fmaxf(to_cam.length2(), 1e-9f);
        float dist = sqrtf(d2);
        Vec3 wc = to_cam * (1.f / dist);
        float cos_c = fmaxf(1e-6f, (-wc).dot(cam_fwd));
        Ray sh(fmadd(wc, EPSILON, pos), wc);
        if (scene_occluded(sv, sh, dist - 2.f * EPSILON, tc))
            return;
        float W = sv.cam.focal * sv.cam.focal / (cos_c * cos_c * cos_c);
        Vec3 val = f_times_cos * (W / d2) * inv_npix * caustic_scaling;
        if (!val.has_nan()) splat(py * sv.cam.w + px, val);
    };

    // direct emitter->camera splat (the light itself is visible)
    if (em.type == EM_AREA || em.type == EM_AREA_SPOT) {
        Vec3 to_cam = sv.cam.pos - le.ray.o;
        float dist = to_cam.length();
        Vec3 wc = to_cam * (1.f / fmaxf(dist, 1e-9f));
        Vec3 l_emit = emitter_eval_le(em, le.normal, wc, le.uv, sv.textures);
        // emission splat uses area pdf only: beta = Le cos / (inv_area) -> here
        // we re-derive from emission directly: Le * cos(n, wc) / (pdf_pos * epdf)
        float cos_l = le.normal.dot(wc);
        if (cos_l > 0.f && !l_emit.is_zero())
            connect(le.ray.o, l_emit * (cos_l / fmaxf(em.inv_area * epdf, 1e-12f)), 0);
    }

    Vec3 thp = le.throughput / epdf;
    Ray ray = le.ray;
    int n_spec = 0;
    PathStats st;
    int b = 0;
    for (int iter = 0; iter < sv.md.max_depth * 2 + 8 && b < sv.md.max_depth; ++iter) {
        HitRecord hit = scene_intersect(sv, ray, MAX_DIST, tc);
        if (hit.prim_idx < 0) break;
        Vec3 pos = ray.at(hit.t);
        uint32_t po = sv.prim_obj[hit.prim_idx];
        bool is_sphere = (po & PRIM_SPHERE_BIT) != 0;
        const ObjInfo& obj = sv.objs[po & PRIM_OBJ_MASK];
        const Prim prim = sv.prims[hit.prim_idx];
        Interaction it = get_interaction(prim, sv.attrs[hit.prim_idx], is_sphere, pos, hit.u, hit.v);
        const BsdfParams& bsdf = sv.bsdfs[obj.bsdf_id];

        // connect this vertex to the camera through the (adjoint) BSDF
        if (!bsdf_is_delta(bsdf)) {
            Vec3 to_cam = (sv.cam.pos - pos).normalized();
            Vec3 f = bsdf_eval(bsdf, -ray.d, to_cam, it, sv.textures);  // includes |cos|
            connect(pos, thp * f, n_spec);
        }

        BsdfSample bs = bsdf_sample(bsdf, -ray.d, it, sp, sv.textures, &path_lambda);
        if (bs.pdf <= 0.f || bs.weight.is_zero() || bs.weight.has_nan() || bs.wi.has_nan()) break;
        thp *= bs.weight;
        if (bs.lobe & (LOBE_SPECULAR | LOBE_TRANSMIT)) ++n_spec;
        if (!(bs.lobe & LOBE_NULL)) {
            if (bs.lobe & LOBE_DIFFUSE)  { if (++st.n_diffuse  > sv.md.max_diffuse)  break; }
            if (bs.lobe & LOBE_SPECULAR) { if (++st.n_specular > sv.md.max_specular) break; }
            if (bs.lobe & LOBE_TRANSMIT) { if (++st.n_transmit > sv.md.max_transmit) break; }
            ++b;
        }
        ray = Ray(fmadd(bs.wi, EPSILON, pos), bs.wi);
        if (b > 1) {
            float p = clampv(thp.max_elem(), 0.f, 1.f);
            if (p < 0.1f) {
                if (sp.next1f() >= p * 10.f) break;
                thp *= (1.f / (p * 10.f));
            }
        }
    }
}

// Host flavor: splat into a plain float RGBA accumulator.
inline void trace_light_path(const SceneView& sv, Sampler& sp, float* img, int w, int h,
                             int spec_constraint, float caustic_scaling) {
    trace_light_path_impl(sv, sp, [&](int pix, Vec3 v) {
        img[pix * 4 + 0] += v.x;
        img[pix * 4 + 1] += v.y;
        img[pix * 4 + 2] += v.z;
    }, TravCtx{}, spec_constraint, caustic_scaling);
}

} // namespace hippt
