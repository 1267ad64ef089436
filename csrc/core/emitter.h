// emitter.h — emitters as tagged unions: point, area, area-spot, envmap.
//
// Capability parity: reference src/core/emitter.cuh (PointSource, AreaSource
// with uniform surface sampling + solid-angle pdf conversion + optional
// emission texture, AreaSpotSource cone restriction, EnvMapEmitter lat-long
// HDRI with azimuth/zenith rotation) + impl/emitter.cu.
//
// MI355X-native change: area emitters sample their primitive AREA-WEIGHTED
// through a per-emitter CDF built on the host (reference picks uniformly via
// CompactedObjInfo::sample_emitter_primitive); pdf is exactly 1/total_area.
#pragma once
#include "geometry.h"
#include "texture.h"
#include "rng.h"
#include "sampling.h"

namespace hippt {

enum EmitterType : int {
    EM_NONE = 0,
    EM_POINT,
    EM_AREA,
    EM_AREA_SPOT,
    EM_ENVMAP,
};

struct alignas(16) EmitterParams {
    Vec4 emission;     // radiance (area/envmap scale) or intensity (point); w = scale
    Vec4 aux;          // point: xyz=pos; spot: xyz=axis?, w=cos_max; envmap: x=azimuth,y=zenith
    int32_t type;
    int32_t obj_id;    // bound object (area) or -1
    int32_t tex_id;    // emission texture (area uv / envmap latlong) or -1
    int32_t prim_base; // offset into emitter_prims / emitter_cdf
    int32_t prim_cnt;
    float inv_area;    // 1 / total surface area (area emitters)
    float extra0, extra1;
};

struct EmitterSampleRec {
    Vec3 pos;       // point on the light (for envmap: far point)
    Vec3 normal;    // light-side normal
    Vec3 radiance;  // emitted radiance toward the shading point (point: I/r^2 premult)
    float pdf;      // solid-angle pdf as seen from the shading point (delta: 1)
    bool delta;
};

// geometry arrays needed to sample area lights
struct EmitterGeom {
    const Prim* prims;
    const PrimAttr* attrs;
    const uint32_t* prim_obj;
    const int* emitter_prims;     // BVH-reordered primitive ids per emitter
    const float* emitter_cdf;     // per-emitter cumulative areas, normalized [0,1]
    const TexView* textures;
    // envmap importance sampling (beyond reference: the reference only does
    // cosine-hemisphere NEE, emitter.cu:25-73): row-marginal and per-row
    // conditional CDFs over luminance x sin(theta); null = cosine fallback
    const float* env_rows;        // (h) cumulative
    const float* env_cols;        // (h*w) cumulative per row
    int env_w, env_h;
    // scene bounding sphere (envmap light-path emission: disk sampling)
    Vec3 scene_center;
    float scene_radius;
};

// first index with cdf[i] > u (cdf ascending, cdf[n-1] == 1)
HD int cdf_find(const float* cdf, int n, float u) {
    int lo = 0, hi = n - 1;
    while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (cdf[mid] > u) hi = mid;
        else lo = mid + 1;
    }
    return lo;
}

HD Vec3 emitter_radiance_tex(const EmitterParams& e, const TexView* textures, Vec2 uv) {
    Vec3 rad = e.emission.xyz() * e.emission.w;
    if (e.tex_id >= 0) rad *= textures[e.tex_id].sample(uv).xyz();
    return rad;
}

// Evaluate radiance leaving light surface point toward direction `to_viewer`
// (unit, points away from the light surface). Area-type only.
HD Vec3 emitter_eval_le(const EmitterParams& e, const Vec3& light_n, const Vec3& to_viewer,
                        Vec2 uv, const TexView* textures) {
    float c = light_n.dot(to_viewer);
    if (c <= 0.f) return Vec3(0.f);
    if (e.type == EM_AREA_SPOT) {
        // cone restriction around the surface normal (emitter.cuh:225-311)
        if (c < e.aux.w) return Vec3(0.f);
    }
    return emitter_radiance_tex(e, textures, uv);
}

// Envmap lookup for a world direction (ray miss path).
HD Vec3 envmap_eval(const EmitterParams& e, const Vec3& dir, const TexView* textures) {
    Vec3 d = dir;
    float az = e.aux.x, ze = e.aux.y;
    if (az != 0.f || ze != 0.f) {
        Quat q = Quat::angle_axis(-az, Vec3(0.f, 1.f, 0.f)) * Quat::angle_axis(-ze, Vec3(1.f, 0.f, 0.f));
        d = q.rotate(d);
    }
    Vec3 rad = e.emission.xyz() * e.emission.w;
    if (e.tex_id >= 0) rad *= textures[e.tex_id].sample(dir_to_latlong(d)).xyz();
    return rad;
}

// Importance-sample a world direction TOWARD the environment from its
// luminance CDF; solid-angle pdf out.  Shared by NEE (emitter_sample) and
// light-path emission (emitter_sample_le).  Requires g.env_rows + texture.
HD Vec3 envmap_sample_dir(const EmitterParams& e, const EmitterGeom& g,
                          Sampler& sampler, float& pdf) {
    const int w = g.env_w, h = g.env_h;
    int row = cdf_find(g.env_rows, h, sampler.next1f());
    const float* crow = g.env_cols + (size_t)row * w;
    int col = cdf_find(crow, w, sampler.next1f());
    float pr = g.env_rows[row] - (row ? g.env_rows[row - 1] : 0.f);
    float pc = crow[col] - (col ? crow[col - 1] : 0.f);
    float uu = (col + sampler.next1f()) / w;
    float vv = (row + sampler.next1f()) / h;
    float theta = vv * PI, phi = (uu - 0.5f) * 2.f * PI;
    float st = sinf(theta);
    Vec3 dl(st * sinf(phi), cosf(theta), -st * cosf(phi));
    // invert the envmap rotation applied in envmap_eval
    Vec3 dir = dl;
    float az = e.aux.x, ze = e.aux.y;
    if (az != 0.f || ze != 0.f) {
        Quat qi = Quat::angle_axis(ze, Vec3(1.f, 0.f, 0.f)) *
                  Quat::angle_axis(az, Vec3(0.f, 1.f, 0.f));
        dir = qi.rotate(dl);
    }
    pdf = pr * pc * (float)w * (float)h / (2.f * PI * PI * fmaxf(st, 1e-5f));
    return dir;
}

// NEE sample toward emitter e from shading point `sp_pos` with normal `sp_n`.
HD EmitterSampleRec emitter_sample(const EmitterParams& e, const EmitterGeom& g,
                                   const Vec3& sp_pos, const Vec3& sp_n, Sampler& sampler) {
    EmitterSampleRec r{};
    switch (e.type) {
    case EM_POINT: {
        r.pos = e.aux.xyz();
        Vec3 d = r.pos - sp_pos;
        float d2 = fmaxf(d.length2(), 1e-8f);
        r.normal = (sp_pos - r.pos) * (1.f / sqrtf(d2));
        r.radiance = e.emission.xyz() * (e.emission.w / d2);  // 1/r^2 attenuation
        r.pdf = 1.f;
        r.delta = true;
        return r;
    }
    case EM_AREA:
    case EM_AREA_SPOT: {
        // area-weighted primitive pick via CDF
        int cnt = e.prim_cnt;
        if (cnt <= 0) { r.pdf = 0.f; return r; }
        float xi = sampler.next1f();
        const float* cdf = g.emitter_cdf + e.prim_base;
        int lo = 0, hi = cnt - 1;
        while (lo < hi) { int mid = (lo + hi) >> 1; if (cdf[mid] < xi) lo = mid + 1; else hi = mid; }
        int pid = g.emitter_prims[e.prim_base + lo];
        const Prim p = g.prims[pid];
        bool sph = (g.prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
        Vec3 lpos, ln; Vec2 uv{0.f, 0.f};
        if (sph) {
            float pdf_dir;
            Vec3 dir = sample_uniform_sphere(sampler.next2f(), pdf_dir);
            lpos = p.v0.xyz() + dir * p.v0.w;
            ln = dir;
        } else {
            Vec3 bc = sample_triangle_bary(sampler.next2f());
            lpos = p.v0.xyz() + p.e1.xyz() * bc.y + p.e2.xyz() * bc.z;
            const PrimAttr a = g.attrs[pid];
            Vec3 nsum = a.n0.xyz() * bc.x + a.n1.xyz() * bc.y + a.n2.xyz() * bc.z;
            ln = nsum.length2() > 1e-16f ? nsum.normalized()
                                         : p.e1.xyz().cross(p.e2.xyz()).normalized();
            Vec2 uv0{a.n0.w, a.n1.w}, uv1{a.n2.w, a.uvrest.x}, uv2{a.uvrest.y, a.uvrest.z};
            uv = uv0 * bc.x + uv1 * bc.y + uv2 * bc.z;
        }
        Vec3 to_sp = sp_pos - lpos;
        float d2 = fmaxf(to_sp.length2(), 1e-9f);
        float dist = sqrtf(d2);
        Vec3 wo_light = to_sp * (1.f / dist);
        float cos_l = ln.dot(wo_light);
        r.pos = lpos;
        r.normal = ln;
        r.radiance = emitter_eval_le(e, ln, wo_light, uv, g.textures);
        // pdf_area = inv_area (area-weighted) -> solid angle
        r.pdf = cos_l > 1e-6f ? e.inv_area * d2 / cos_l : 0.f;
        r.delta = false;
        return r;
    }
    case EM_ENVMAP: {
        if (g.env_rows && e.tex_id >= 0) {
            // luminance-CDF importance sampling over the lat-long texture
            float pdf;
            Vec3 dir = envmap_sample_dir(e, g, sampler, pdf);
            r.pos = sp_pos + dir * ENVMAP_DIST;
            r.normal = -dir;
            r.radiance = envmap_eval(e, dir, g.textures);
            r.pdf = pdf;
            r.delta = false;
            return r;
        }
        // cosine-hemisphere NEE around the shading normal (emitter.cu:25-73)
        float pdf;
        Vec3 nn = sp_n;
        Vec3 local = sample_cosine_hemisphere(sampler.next2f(), pdf);
        Vec3 dir = Frame::from_n(nn).to_world(local);
        r.pos = sp_pos + dir * ENVMAP_DIST;
        r.normal = -dir;
        r.radiance = envmap_eval(e, dir, g.textures);
        r.pdf = pdf;
        r.delta = false;
        return r;
    }
    default:
        r.pdf = 0.f;
        return r;
    }
}

// Solid-angle pdf of hitting this (area/env) emitter with a BSDF ray, for MIS.
HD float emitter_pdf_hit(const EmitterParams& e, const Vec3& dir, float dist,
                         const Vec3& light_n, const Vec3& sp_n,
                         const EmitterGeom& g) {
    if (e.type == EM_AREA || e.type == EM_AREA_SPOT) {
        float cos_l = light_n.dot(-dir);
        if (cos_l <= 1e-6f) return 0.f;
        return e.inv_area * dist * dist / cos_l;
    }
    if (e.type == EM_ENVMAP) {
        if (g.env_rows && e.tex_id >= 0) {
            // must mirror the tabulated NEE pdf exactly (MIS)
            Vec3 d = dir;
            float az = e.aux.x, ze = e.aux.y;
            if (az != 0.f || ze != 0.f) {
                Quat q = Quat::angle_axis(-az, Vec3(0.f, 1.f, 0.f)) *
                         Quat::angle_axis(-ze, Vec3(1.f, 0.f, 0.f));
                d = q.rotate(d);
            }
            Vec2 uv = dir_to_latlong(d);
            const int w = g.env_w, h = g.env_h;
            int col = (int)(uv.x * w);
            int row = (int)(uv.y * h);
            col = col < 0 ? 0 : (col >= w ? w - 1 : col);
            row = row < 0 ? 0 : (row >= h ? h - 1 : row);
            const float* crow = g.env_cols + (size_t)row * w;
            float pr = g.env_rows[row] - (row ? g.env_rows[row - 1] : 0.f);
            float pc = crow[col] - (col ? crow[col - 1] : 0.f);
            float st = sinf(uv.y * PI);
            return pr * pc * (float)w * (float)h /
                   (2.f * PI * PI * fmaxf(st, 1e-5f));
        }
        float c = sp_n.dot(dir);
        return c > 0.f ? c * INV_PI : 0.f;
    }
    return 0.f;
}

// Light-tracing emission sample: position + direction + power weight.
// (reference Emitter::sample_le; used by the LightTracer)
struct EmitterLeRec {
    Ray ray;
    Vec3 normal;
    Vec3 throughput;  // radiance * cos / (pdf_pos * pdf_dir), ready to trace
    Vec2 uv;
    bool valid;
};

HD EmitterLeRec emitter_sample_le(const EmitterParams& e, const EmitterGeom& g, Sampler& sampler) {
    EmitterLeRec r{};
    switch (e.type) {
    case EM_POINT: {
        float pdf_dir;
        Vec3 d = sample_uniform_sphere(sampler.next2f(), pdf_dir);
        r.ray = Ray(e.aux.xyz(), d);
        r.normal = d;
        r.throughput = e.emission.xyz() * (e.emission.w / pdf_dir);
        r.valid = true;
        return r;
    }
    case EM_AREA:
    case EM_AREA_SPOT: {
        int cnt = e.prim_cnt;
        if (cnt <= 0) return r;
        float xi = sampler.next1f();
        const float* cdf = g.emitter_cdf + e.prim_base;
        int lo = 0, hi = cnt - 1;
        while (lo < hi) { int mid = (lo + hi) >> 1; if (cdf[mid] < xi) lo = mid + 1; else hi = mid; }
        int pid = g.emitter_prims[e.prim_base + lo];
        const Prim p = g.prims[pid];
        bool sph = (g.prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
        Vec3 lpos, ln; Vec2 uv{0.f, 0.f};
        if (sph) {
            float pdf_sph;
            Vec3 dirn = sample_uniform_sphere(sampler.next2f(), pdf_sph);
            lpos = p.v0.xyz() + dirn * p.v0.w;
            ln = dirn;
        } else {
            Vec3 bc = sample_triangle_bary(sampler.next2f());
            lpos = p.v0.xyz() + p.e1.xyz() * bc.y + p.e2.xyz() * bc.z;
            const PrimAttr a = g.attrs[pid];
            Vec3 nsum = a.n0.xyz() * bc.x + a.n1.xyz() * bc.y + a.n2.xyz() * bc.z;
            ln = nsum.length2() > 1e-16f ? nsum.normalized()
                                         : p.e1.xyz().cross(p.e2.xyz()).normalized();
        }
        float pdf_dir;
        Vec3 local = (e.type == EM_AREA_SPOT)
                         ? sample_uniform_cone(sampler.next2f(), e.aux.w, pdf_dir)
                         : sample_cosine_hemisphere(sampler.next2f(), pdf_dir);
        Vec3 dir = Frame::from_n(ln).to_world(local);
        float cos_l = ln.dot(dir);
        if (cos_l <= 0.f || pdf_dir <= 0.f) return r;
        r.ray = Ray(fmadd(ln, EPSILON, lpos), dir);
        r.normal = ln;
        r.uv = uv;
        Vec3 rad = emitter_radiance_tex(e, g.textures, uv);
        // pdf_pos = inv_area; throughput = L * cos / (pdf_pos * pdf_dir)
        r.throughput = rad * (cos_l / fmaxf(e.inv_area * pdf_dir, 1e-12f));
        r.valid = true;
        return r;
    }
    case EM_ENVMAP: {
        // Light-path emission from the environment (reference
        // EnvMapEmitter::sample_le, emitter.cuh:338): pick a direction toward
        // the env (importance-sampled from the luminance CDF when present,
        // else uniform sphere), then a point on the disk of the scene
        // bounding sphere perpendicular to it, and shoot the ray INWARD.
        // pdf_pos = 1/(pi R^2) over the disk; the disk is perpendicular to
        // the ray so the cosine is 1:
        //   throughput = Le(dir) * pi R^2 / pdf_dir.
        float pdf_dir;
        Vec3 dir;  // from scene toward env
        if (g.env_rows && e.tex_id >= 0) {
            dir = envmap_sample_dir(e, g, sampler, pdf_dir);
        } else {
            dir = sample_uniform_sphere(sampler.next2f(), pdf_dir);
        }
        if (pdf_dir <= 0.f) return r;
        float R = fmaxf(g.scene_radius, 1e-4f);
        Vec2 u = sampler.next2f();
        float rr = sqrtf(u.x), phi = 2.f * PI * u.y;
        Frame f = Frame::from_n(dir);
        Vec3 offset = f.to_world(Vec3(rr * cosf(phi), rr * sinf(phi), 0.f)) * R;
        Vec3 origin = g.scene_center + dir * (2.f * R) + offset;
        r.ray = Ray(origin, -dir);
        r.normal = -dir;
        r.throughput = envmap_eval(e, dir, g.textures) * (PI * R * R / pdf_dir);
        r.valid = true;
        return r;
    }
    default:
        return r;
    }
}

} // namespace hippt
