// integrator_vol.h — volumetric unidirectional path tracing with nested-media
// stack, medium distance sampling, transmittance NEE and medium emission.
//
// Capability parity: reference src/pt_impl/megakernel_vpt.cu (render_vpt_kernel
// :204-456: per-thread nested-volume BankStack :64-94, alpha-masked forward
// boundary skip :269-284, occlusion_transmittance_estimate NEE :104-201,
// medium emission via query_emission, ToF min/max gating :48-57).
#pragma once
#include "integrator.h"

namespace hippt {

// 4-deep nested-media stack (reference BankStack: 4 x u8 in one uchar4).
struct VolStack {
    int8_t s[4];
    int top;  // number of entries
    HD VolStack() : top(0) { s[0] = s[1] = s[2] = s[3] = -1; }
    HD void push(int m) { if (top < 4) s[top++] = (int8_t)m; }
    HD void pop() { if (top > 0) --top; }
    HD int current() const { return top > 0 ? s[top - 1] : -1; }
};

// Does this surface pass light through without shading? (null boundary)
HD bool is_null_boundary(const SceneView& sv, const ObjInfo& obj) {
    return (obj.flags & OBJ_CULLABLE) != 0 || sv.bsdfs[obj.bsdf_id].type == BSDF_FORWARD;
}

// Update the media stack when crossing a boundary along direction d.
HD void cross_boundary(VolStack& st, const ObjInfo& obj, const Vec3& d, const Vec3& geo_n) {
    bool entering = d.dot(geo_n) < 0.f;
    if (entering) { if (obj.medium_in >= 0) st.push(obj.medium_in); }
    else {
        if (st.current() == obj.medium_in && obj.medium_in >= 0) st.pop();
        else if (st.top > 0) st.pop();
    }
}

// Transmittance estimate along a shadow path (re-trace across null interfaces
// accumulating per-segment medium transmittance; reference
// occlusion_transmittance_estimate, megakernel_vpt.cu:104-201).
HD Vec3 transmittance_estimate(const SceneView& sv, Vec3 from, const Vec3& wi, float dist,
                               TravCtx tc,
                               VolStack stack, Sampler& sp) {
    Vec3 tr(1.f);
    float remaining = dist;
    for (int hop = 0; hop < 32; ++hop) {
        Ray r(fmadd(wi, EPSILON, from), wi);
        remaining -= EPSILON;
        if (remaining <= EPSILON) break;
        HitRecord hit = scene_intersect(sv, r, remaining, tc);
        float seg = hit.prim_idx >= 0 ? hit.t : remaining;
        int med = stack.current();
        if (med >= 0) {
            tr *= medium_transmittance(sv.media[med], r, seg, sp);
            if (tr.max_elem() < 1e-6f) return Vec3(0.f);
        }
        if (hit.prim_idx < 0) break;  // reached the light
        uint32_t po = sv.prim_obj[hit.prim_idx];
        const ObjInfo& obj = sv.objs[po & PRIM_OBJ_MASK];
        if (!is_null_boundary(sv, obj)) return Vec3(0.f);  // opaque blocker
        Vec3 hp = r.at(hit.t);
        Vec3 gn = geometric_normal(sv.prims[hit.prim_idx], (po & PRIM_SPHERE_BIT) != 0, hp);
        cross_boundary(stack, obj, wi, gn);
        from = hp;
        remaining -= hit.t;
    }
    return tr;
}

HD Vec3 trace_path_volumetric(const SceneView& sv, Ray ray, Sampler& sp, TravCtx tc = {}) {
    float path_lambda = 0.f;
    Vec3 L(0.f), thp(1.f);
    float prev_pdf = 0.f;
    bool prev_delta = true;
    Vec3 prev_n(0.f, 0.f, 1.f);
    float path_time = 0.f;
    PathStats st;
    VolStack stack;
    if (sv.cam_medium >= 0) stack.push(sv.cam_medium);

    int b = 0;
    for (int iter = 0; iter < sv.md.max_depth * 3 + 16 && b < sv.md.max_depth; ++iter) {
        HitRecord hit = scene_intersect(sv, ray, MAX_DIST, tc);
        float t_surf = hit.prim_idx >= 0 ? hit.t : MAX_DIST;

        // ---- medium flight
        int med = stack.current();
        bool scattered = false;
        float t_event = t_surf;
        if (med >= 0) {
            MediumSample ms = medium_sample(sv.media[med], ray, t_surf, sp);
            thp *= ms.local_thp;
            if (thp.is_zero() || thp.has_nan()) break;
            if (ms.scattered) { scattered = true; t_event = ms.dist; }
        }

        if (scattered) {
            // ---- volume scattering event
            Vec3 pos = ray.at(t_event);
            path_time += t_event;
            const MediumParams& mp = sv.media[med];
            const PhaseParams& ph = sv.phases[mp.phase_id];
            // medium emission (blackbody, query_emission parity)
            Vec3 em = medium_emission(mp, pos, sp);
            if (!em.is_zero() && tof_in_range(sv.md, path_time)) L += thp * em;
            // NEE from the volume point
            if (sv.n_emitters > 0) {
                float epdf;
                int ei = pick_emitter(sv, sp, epdf);
                EmitterSampleRec er = emitter_sample(sv.emitters[ei], sv.emitter_geom(), pos,
                                                     ray.d, sp);
                if (er.pdf > 0.f && !er.radiance.is_zero()) {
                    Vec3 to_l = er.pos - pos;
                    float dist = to_l.length();
                    Vec3 wi = to_l * (1.f / fmaxf(dist, 1e-9f));
                    float fp = phase_eval(ph, ray.d.dot(wi));
                    Vec3 tr = transmittance_estimate(sv, pos, wi, dist - EPSILON, tc, stack, sp);
                    if (!tr.is_zero()) {
                        float light_pdf = er.pdf * epdf;
                        float w = er.delta ? 1.f : mis_weight(light_pdf, fp);
                        if (tof_in_range(sv.md, path_time + dist))
                            L += thp * tr * er.radiance * (fp * w / light_pdf);
                    }
                }
            }
            // phase scatter
            PhaseSampleRec ps = phase_sample(ph, ray.d, sp);
            thp *= ps.weight;
            prev_pdf = ps.pdf;
            prev_delta = false;
            prev_n = ray.d;
            ray = Ray(pos, ps.wi);
            if (++st.n_volume > sv.md.max_volume) break;
            ++b;
        } else if (hit.prim_idx >= 0) {
            // ---- surface event
            Vec3 pos = ray.at(hit.t);
            path_time += hit.t;
            uint32_t po = sv.prim_obj[hit.prim_idx];
            bool is_sphere = (po & PRIM_SPHERE_BIT) != 0;
            const ObjInfo& obj = sv.objs[po & PRIM_OBJ_MASK];
            const Prim prim = sv.prims[hit.prim_idx];
            Vec3 geo_n = geometric_normal(prim, is_sphere, pos);

            if (is_null_boundary(sv, obj)) {
                // pass through, flip media stack (megakernel_vpt.cu:269-284)
                cross_boundary(stack, obj, ray.d, geo_n);
                ray = Ray(fmadd(ray.d, EPSILON, pos), ray.d);
                continue;  // does not count as a bounce
            }

            Interaction it = get_interaction(prim, sv.attrs[hit.prim_idx], is_sphere, pos, hit.u, hit.v);
            const BsdfParams& bsdf = sv.bsdfs[obj.bsdf_id];
            if (bsdf.tex[TEX_NORMAL] >= 0)
                it.shading_n = apply_normal_map(sv.textures, bsdf.tex[TEX_NORMAL], it.uv, it.shading_n);

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

            if (!bsdf_is_delta(bsdf) && sv.n_emitters > 0) {
                float epdf;
                int ei = pick_emitter(sv, sp, epdf);
                EmitterSampleRec er = emitter_sample(sv.emitters[ei], sv.emitter_geom(), pos,
                                                     it.shading_n, sp);
                if (er.pdf > 0.f && !er.radiance.is_zero()) {
                    Vec3 to_l = er.pos - pos;
                    float dist = to_l.length();
                    Vec3 wi = to_l * (1.f / fmaxf(dist, 1e-9f));
                    Vec3 f = bsdf_eval(bsdf, -ray.d, wi, it, sv.textures);
                    if (!f.is_zero()) {
                        float sh_max = (sv.emitters[ei].type == EM_ENVMAP ? ENVMAP_DIST : dist) - 2.f * EPSILON;
                        Vec3 tr = transmittance_estimate(sv, pos, wi, sh_max, tc, stack, sp);
                        if (!tr.is_zero()) {
                            float light_pdf = er.pdf * epdf;
                            float w = er.delta ? 1.f
                                               : mis_weight(light_pdf, bsdf_pdf(bsdf, -ray.d, wi, it, sv.textures));
                            if (tof_in_range(sv.md, path_time + dist))
                                L += thp * tr * f * er.radiance * (w / light_pdf);
                        }
                    }
                }
            }

            BsdfSample bs = bsdf_sample(bsdf, -ray.d, it, sp, sv.textures, &path_lambda);
            if (bs.pdf <= 0.f || bs.weight.is_zero() || bs.weight.has_nan() || bs.wi.has_nan()) break;
            thp *= bs.weight;
            if ((bs.lobe & LOBE_TRANSMIT) != 0) cross_boundary(stack, obj, bs.wi, geo_n);
            if (!(bs.lobe & LOBE_NULL)) {
                if (bs.lobe & LOBE_DIFFUSE)  { if (++st.n_diffuse  > sv.md.max_diffuse)  break; }
                if (bs.lobe & LOBE_SPECULAR) { if (++st.n_specular > sv.md.max_specular) break; }
                if (bs.lobe & LOBE_TRANSMIT) { if (++st.n_transmit > sv.md.max_transmit) break; }
                ++b;
            }
            prev_delta = (bs.lobe & LOBE_DELTA) != 0;
            prev_pdf = bs.pdf;
            prev_n = it.shading_n;
            ray = Ray(fmadd(bs.wi, EPSILON, pos), bs.wi);
        } else {
            // ---- miss -> envmap
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
            break;
        }

        // Russian roulette
        if (b > 1) {
            float p = clampv(thp.max_elem(), 0.f, 1.f);
            if (p < 0.1f) {
                if (sp.next1f() >= p * 10.f) break;
                thp *= (1.f / (p * 10.f));
            }
        }
    }
    if (L.has_nan()) return Vec3(0.f);
    return L;
}

} // namespace hippt
