// bsdf.h — the 8-type BSDF system as a tagged union with switch dispatch.
//
// Capability parity: reference src/bsdf/* (Lambertian, Specular mirror,
// Translucent glass, Plastic, PlasticForward, GGX anisotropic conductor,
// Dispersion spectral glass, Forward null) + preset tables
// (src/core/preset_params.cuh).  The reference realizes polymorphism with
// on-device virtual classes created by 1-thread kernels; on MI355X (wave64,
// divergence-hostile) we use POD parameter blocks + enum switch — same
// feature matrix, no device vtables (SURVEY.md L2 note).
//
// Conventions:
//   wo = unit vector from the hit point toward the previous vertex (viewer),
//   wi = unit vector toward the next vertex (sampled / light direction),
//   both in world space.  eval() returns f(wo,wi) * |cos(n,wi)|; delta lobes
//   eval to 0 and pdf to 0 and are only reachable through sample().
#pragma once
#include "fresnel.h"
#include "frame.h"
#include "sampling.h"
#include "texture.h"
#include "spectrum.h"
#include "rng.h"
#include "geometry.h"

namespace hippt {

enum BsdfType : int {
    BSDF_LAMBERTIAN = 0,
    BSDF_SPECULAR,
    BSDF_TRANSLUCENT,
    BSDF_PLASTIC,
    BSDF_PLASTIC_FORWARD,
    BSDF_GGX_CONDUCTOR,
    BSDF_DISPERSION,
    BSDF_FORWARD,
    BSDF_NTYPES
};

// scatter lobe tags (per-lobe bounce caps; reference max_depth.h)
enum : uint32_t {
    LOBE_DIFFUSE  = 1u,
    LOBE_SPECULAR = 2u,   // counts against max_specular
    LOBE_TRANSMIT = 4u,
    LOBE_GLOSSY   = 8u,
    LOBE_DELTA    = 16u,  // dirac lobe -> NEE skipped, MIS weight 1 on hit
    LOBE_NULL     = 32u,  // forward pass-through (volume boundary)
};

struct alignas(16) BsdfParams {
    Vec4 kd, ks, kg;          // type-interpreted parameter triplet (reference bsdf.cuh)
    int32_t type;
    float ior;                // dielectric IOR (translucent/plastic)
    float extra0, extra1;     // ggx: roughness_x/y; plastic: trans_scaler/thickness; dispersion: cauchy A/B
    int16_t tex[8];           // TEX_DIFFUSE..TEX_ROUGHNESS slots, -1 = none

    HOSTFN static BsdfParams make(int t) {
        BsdfParams b{};
        b.kd = Vec4(0.8f, 0.8f, 0.8f, 0.f);
        b.ks = Vec4(1.f, 1.f, 1.f, 0.f);
        b.kg = Vec4(0.f, 0.f, 0.f, 0.f);
        b.type = t; b.ior = 1.5f; b.extra0 = 0.f; b.extra1 = 0.f;
        for (int i = 0; i < 8; ++i) b.tex[i] = -1;
        return b;
    }
};

struct BsdfSample {
    Vec3 wi;
    Vec3 weight;   // f * |cos| / pdf  (full throughput multiplier)
    float pdf;     // solid-angle pdf of the sampled direction (1 for delta)
    uint32_t lobe;
};

HD float roughness_to_alpha(float r) { return fmaxf(1e-4f, r * r); }

// ---------------------------------------------------------------- GGX terms
HD float ggx_d(const Vec3& wh_l, float ax, float ay) {
    float t = wh_l.x * wh_l.x / (ax * ax) + wh_l.y * wh_l.y / (ay * ay) + wh_l.z * wh_l.z;
    return 1.f / (PI * ax * ay * t * t);
}
HD float ggx_lambda(const Vec3& w_l, float ax, float ay) {
    float c2 = w_l.z * w_l.z;
    if (c2 >= 1.f) return 0.f;
    float a2 = (w_l.x * w_l.x * ax * ax + w_l.y * w_l.y * ay * ay) / fmaxf(c2, 1e-12f);
    return 0.5f * (-1.f + sqrtf(1.f + a2));
}
HD float ggx_g(const Vec3& wo_l, const Vec3& wi_l, float ax, float ay) {
    return 1.f / (1.f + ggx_lambda(wo_l, ax, ay) + ggx_lambda(wi_l, ax, ay));
}
// Heitz 2018 visible-normal sampling (parity: bsdf_ggx.cu:127-145 slope-space VNDF)
HD Vec3 ggx_sample_wh(const Vec3& wo_l, float ax, float ay, Vec2 u) {
    Vec3 vh = Vec3(ax * wo_l.x, ay * wo_l.y, wo_l.z).normalized();
    float lensq = vh.x * vh.x + vh.y * vh.y;
    Vec3 T1 = lensq > 1e-12f ? Vec3(-vh.y, vh.x, 0.f) * (1.f / sqrtf(lensq)) : Vec3(1.f, 0.f, 0.f);
    Vec3 T2 = vh.cross(T1);
    float r = sqrtf(u.x);
    float phi = TWO_PI * u.y;
    float t1 = r * cosf(phi), t2 = r * sinf(phi);
    float s = 0.5f * (1.f + vh.z);
    t2 = (1.f - s) * sqrtf(fmaxf(0.f, 1.f - t1 * t1)) + s * t2;
    Vec3 nh = T1 * t1 + T2 * t2 + vh * sqrtf(fmaxf(0.f, 1.f - t1 * t1 - t2 * t2));
    return Vec3(ax * nh.x, ay * nh.y, fmaxf(1e-6f, nh.z)).normalized();
}
HD float ggx_pdf_vndf(const Vec3& wo_l, const Vec3& wh_l, float ax, float ay) {
    float g1 = 1.f / (1.f + ggx_lambda(wo_l, ax, ay));
    return g1 * fabsf(wo_l.dot(wh_l)) * ggx_d(wh_l, ax, ay) / fmaxf(fabsf(wo_l.z), 1e-7f);
}

// ------------------------------------------------------------ helper: albedo
HD Vec3 bsdf_albedo(const BsdfParams& b, Vec2 uv, const TexView* textures) {
    return tex_or(textures, b.tex[TEX_DIFFUSE], uv, b.kd.xyz());
}

// Cauchy IOR: n(lambda) = A + B / lambda_um^2 (reference dispersion.cuh:61-68)
HD float cauchy_ior(float A, float B, float lambda_nm) {
    float lum = lambda_nm * 1e-3f;
    return A + B / (lum * lum);
}

// plastic interlayer absorption along both path legs
HD Vec3 plastic_absorption(const BsdfParams& b, float cos_i, float cos_o) {
    float thickness = b.extra1;
    if (thickness <= 0.f) return Vec3(1.f);
    Vec3 sigma = b.kg.xyz();
    float path = thickness * (1.f / fmaxf(cos_i, 1e-3f) + 1.f / fmaxf(cos_o, 1e-3f));
    return (sigma * -path).expv();
}

// ------------------------------------------------------------------- eval
HD Vec3 bsdf_eval(const BsdfParams& b, const Vec3& wo, const Vec3& wi,
                  const Interaction& it, const TexView* textures) {
    Vec3 n = it.shading_n;
    float cos_o = n.dot(wo), cos_i = n.dot(wi);
    switch (b.type) {
    case BSDF_LAMBERTIAN: {
        if (cos_o <= 0.f || cos_i <= 0.f) {
            // double-sided diffuse: flip when viewing the back face
            if (cos_o < 0.f && cos_i < 0.f) { cos_i = -cos_i; }
            else return Vec3(0.f);
        }
        return bsdf_albedo(b, it.uv, textures) * (INV_PI * cos_i);
    }
    case BSDF_PLASTIC: {
        if (cos_o <= 0.f || cos_i <= 0.f) return Vec3(0.f);
        float Fo = fresnel_dielectric(cos_o, 1.f, b.ior);
        float Fi = fresnel_dielectric(cos_i, 1.f, b.ior);
        Vec3 diff = bsdf_albedo(b, it.uv, textures);
        Vec3 f = diff * (INV_PI * (1.f - Fo) * (1.f - Fi) * b.extra0) * plastic_absorption(b, cos_i, cos_o);
        return f * cos_i;
    }
    case BSDF_GGX_CONDUCTOR: {
        if (cos_o <= 0.f || cos_i <= 0.f) return Vec3(0.f);
        Frame fr = Frame::from_n(n);
        Vec3 wo_l = fr.to_local(wo), wi_l = fr.to_local(wi);
        Vec3 wh_l = (wo_l + wi_l).normalized();
        float rx = b.extra0, ry = b.extra1;
        // NOTE: a roughness texture forces an ISOTROPIC lobe (one channel
        // drives both axes); anisotropy is constant-parameter only.  Same
        // limitation as the reference's single roughness slot.
        if (b.tex[TEX_ROUGHNESS] >= 0) { rx = ry = textures[b.tex[TEX_ROUGHNESS]].sample(it.uv).x; }
        float ax = roughness_to_alpha(rx), ay = roughness_to_alpha(ry);
        float D = ggx_d(wh_l, ax, ay);
        float G = ggx_g(wo_l, wi_l, ax, ay);
        Vec3 F = fresnel_conductor(wo_l.dot(wh_l), b.kd.xyz(), b.ks.xyz());
        Vec3 tint = tex_or(textures, b.tex[TEX_SPECULAR], it.uv, b.kg.xyz());
        return tint * F * (D * G / (4.f * fmaxf(cos_o, 1e-6f)));  // * cos_i / cos_i cancels
    }
    default:
        return Vec3(0.f);  // delta / null lobes
    }
}

// ------------------------------------------------------------------- pdf
HD float bsdf_pdf(const BsdfParams& b, const Vec3& wo, const Vec3& wi, const Interaction& it,
                  const TexView* textures) {
    Vec3 n = it.shading_n;
    float cos_o = n.dot(wo), cos_i = n.dot(wi);
    switch (b.type) {
    case BSDF_LAMBERTIAN: {
        if (cos_o < 0.f && cos_i < 0.f) return -cos_i * INV_PI;
        if (cos_o <= 0.f || cos_i <= 0.f) return 0.f;
        return cos_i * INV_PI;
    }
    case BSDF_PLASTIC: {
        if (cos_o <= 0.f || cos_i <= 0.f) return 0.f;
        float Fo = fresnel_dielectric(cos_o, 1.f, b.ior);
        return (1.f - Fo) * cos_i * INV_PI;  // specular part is delta
    }
    case BSDF_GGX_CONDUCTOR: {
        if (cos_o <= 0.f || cos_i <= 0.f) return 0.f;
        Frame fr = Frame::from_n(n);
        Vec3 wo_l = fr.to_local(wo), wi_l = fr.to_local(wi);
        Vec3 wh_l = (wo_l + wi_l).normalized();
        float rx = b.extra0, ry = b.extra1;
        // NOTE: a roughness texture forces an ISOTROPIC lobe (one channel
        // drives both axes); anisotropy is constant-parameter only.  Same
        // limitation as the reference's single roughness slot.
        if (b.tex[TEX_ROUGHNESS] >= 0) { rx = ry = textures[b.tex[TEX_ROUGHNESS]].sample(it.uv).x; }
        float ax = roughness_to_alpha(rx), ay = roughness_to_alpha(ry);
        return ggx_pdf_vndf(wo_l, wh_l, ax, ay) / (4.f * fmaxf(fabsf(wo_l.dot(wh_l)), 1e-7f));
    }
    default:
        return 0.f;
    }
}

// ------------------------------------------------------------------- sample
// lambda_io: per-path wavelength slot for spectral dispersion (0 = not yet
// sampled).  The wavelength is sampled ONCE per path at the first dispersive
// transmission and the lambda->RGB basis weight applied exactly once; later
// dispersive events reuse the same wavelength (consistent IOR) with weight 1.
// (Per-event resampling squares the non-white basis expectation: measured
// +36% red in the dispersion furnace.)
HD BsdfSample bsdf_sample(const BsdfParams& b, const Vec3& wo, const Interaction& it,
                          Sampler& sp, const TexView* textures,
                          float* lambda_io = nullptr) {
    BsdfSample s{};
    Vec3 n = it.shading_n;
    float cos_o = n.dot(wo);
    switch (b.type) {
    case BSDF_LAMBERTIAN: {
        Vec3 nn = cos_o < 0.f ? -n : n;   // double-sided
        float pdf;
        Vec3 local = sample_cosine_hemisphere(sp.next2f(), pdf);
        s.wi = Frame::from_n(nn).to_world(local);
        s.pdf = pdf;
        s.weight = bsdf_albedo(b, it.uv, textures);  // f*cos/pdf = albedo
        s.lobe = LOBE_DIFFUSE;
        return s;
    }
    case BSDF_SPECULAR: {
        Vec3 nn = cos_o < 0.f ? -n : n;
        s.wi = reflect_dir(wo, nn);
        s.pdf = 1.f;
        s.weight = tex_or(textures, b.tex[TEX_SPECULAR], it.uv, b.ks.xyz());
        s.lobe = LOBE_SPECULAR | LOBE_DELTA;
        return s;
    }
    case BSDF_TRANSLUCENT: {
        // Fresnel-weighted reflect/refract (reference translucent.cuh:47-97)
        float ior = b.ior;
        if (b.tex[TEX_ROUGHNESS] >= 0) ior = textures[b.tex[TEX_ROUGHNESS]].sample(it.uv).x;
        bool entering = cos_o > 0.f;
        Vec3 nn = entering ? n : -n;
        float eta = entering ? 1.f / ior : ior;  // eta_i/eta_t
        float F = fresnel_dielectric(fabsf(cos_o), entering ? 1.f : ior, entering ? ior : 1.f);
        Vec3 tint = tex_or(textures, b.tex[TEX_SPECULAR], it.uv, b.ks.xyz());
        Vec3 wt;
        bool refr_ok = snell_refraction(wo, nn, eta, wt);
        if (!refr_ok || sp.next1f() < F) {
            s.wi = reflect_dir(wo, nn);
            s.pdf = refr_ok ? F : 1.f;
            s.weight = tint;
            s.lobe = LOBE_SPECULAR | LOBE_DELTA;
        } else {
            s.wi = wt.normalized();
            s.pdf = 1.f - F;
            // radiance transport eta^2 factor (translucent.cuh:47-97)
            s.weight = tint * (eta * eta);
            s.lobe = LOBE_TRANSMIT | LOBE_DELTA;
        }
        return s;
    }
    case BSDF_PLASTIC: {
        Vec3 nn = cos_o < 0.f ? -n : n;
        float aco = fabsf(cos_o);
        float F = fresnel_dielectric(aco, 1.f, b.ior);
        Vec3 tint = tex_or(textures, b.tex[TEX_SPECULAR], it.uv, b.ks.xyz());
        if (sp.next1f() < F) {
            s.wi = reflect_dir(wo, nn);
            s.pdf = F;
            s.weight = tint;
            s.lobe = LOBE_SPECULAR | LOBE_DELTA;
        } else {
            float pdf;
            Vec3 local = sample_cosine_hemisphere(sp.next2f(), pdf);
            s.wi = Frame::from_n(nn).to_world(local);
            float cos_i = fabsf(s.wi.dot(nn));
            float Fi = fresnel_dielectric(cos_i, 1.f, b.ior);
            s.pdf = (1.f - F) * pdf;
            // weight = f*cos/pdf with f = kd/pi (1-Fo)(1-Fi) ts Abs
            Vec3 diff = bsdf_albedo(b, it.uv, textures);
            s.weight = diff * ((1.f - Fi) * b.extra0) * plastic_absorption(b, cos_i, aco);
            s.lobe = LOBE_DIFFUSE;
        }
        return s;
    }
    case BSDF_PLASTIC_FORWARD: {
        // coated delta transmission (reference PlasticForward)
        Vec3 nn = cos_o < 0.f ? -n : n;
        float aco = fabsf(cos_o);
        float F = fresnel_dielectric(aco, 1.f, b.ior);
        Vec3 tint = tex_or(textures, b.tex[TEX_SPECULAR], it.uv, b.ks.xyz());
        if (sp.next1f() < F) {
            s.wi = reflect_dir(wo, nn);
            s.pdf = F;
            s.weight = tint;
            s.lobe = LOBE_SPECULAR | LOBE_DELTA;
        } else {
            s.wi = -wo;
            s.pdf = 1.f - F;
            s.weight = bsdf_albedo(b, it.uv, textures) * b.extra0 * plastic_absorption(b, aco, aco);
            s.lobe = LOBE_TRANSMIT | LOBE_DELTA | LOBE_NULL;
        }
        return s;
    }
    case BSDF_GGX_CONDUCTOR: {
        Vec3 nn = cos_o < 0.f ? -n : n;
        Frame fr = Frame::from_n(nn);
        Vec3 wo_l = fr.to_local(wo);
        float rx = b.extra0, ry = b.extra1;
        // NOTE: a roughness texture forces an ISOTROPIC lobe (one channel
        // drives both axes); anisotropy is constant-parameter only.  Same
        // limitation as the reference's single roughness slot.
        if (b.tex[TEX_ROUGHNESS] >= 0) { rx = ry = textures[b.tex[TEX_ROUGHNESS]].sample(it.uv).x; }
        float ax = roughness_to_alpha(rx), ay = roughness_to_alpha(ry);
        Vec3 wh_l = ggx_sample_wh(wo_l, ax, ay, sp.next2f());
        Vec3 wi_l = reflect_dir(wo_l, wh_l);
        if (wi_l.z <= 0.f) { s.pdf = 0.f; s.weight = Vec3(0.f); s.lobe = LOBE_GLOSSY; return s; }
        float D = ggx_d(wh_l, ax, ay);
        float G = ggx_g(wo_l, wi_l, ax, ay);
        float G1 = 1.f / (1.f + ggx_lambda(wo_l, ax, ay));
        Vec3 F = fresnel_conductor(wo_l.dot(wh_l), b.kd.xyz(), b.ks.xyz());
        Vec3 tint = tex_or(textures, b.tex[TEX_SPECULAR], it.uv, b.kg.xyz());
        s.wi = fr.to_world(wi_l);
        s.pdf = ggx_pdf_vndf(wo_l, wh_l, ax, ay) / (4.f * fmaxf(fabsf(wo_l.dot(wh_l)), 1e-7f));
        // weight = f*cos/pdf simplifies to F * G/G1 * tint with VNDF sampling
        s.weight = tint * F * (G / fmaxf(G1, 1e-7f));
        s.lobe = LOBE_GLOSSY;
        return s;
    }
    case BSDF_DISPERSION: {
        // spectral glass: ONE wavelength per path (reference dispersion.cuh)
        bool have_l = lambda_io && *lambda_io > 0.f;
        float lambda = have_l ? *lambda_io
                              : LAMBDA_MIN + (LAMBDA_MAX - LAMBDA_MIN) * sp.next1f();
        float ior = cauchy_ior(b.extra0, b.extra1, lambda);
        bool entering = cos_o > 0.f;
        Vec3 nn = entering ? n : -n;
        float eta = entering ? 1.f / ior : ior;
        float F = fresnel_dielectric(fabsf(cos_o), entering ? 1.f : ior, entering ? ior : 1.f);
        Vec3 tint = b.ks.xyz();
        Vec3 wt;
        bool refr_ok = snell_refraction(wo, nn, eta, wt);
        if (!refr_ok || sp.next1f() < F) {
            s.wi = reflect_dir(wo, nn);
            s.pdf = refr_ok ? F : 1.f;
            s.weight = tint;  // reflection is not dispersive
            s.lobe = LOBE_SPECULAR | LOBE_DELTA;
        } else {
            Vec3 spectral = have_l ? Vec3(1.f) : wavelength_to_rgb(lambda);
            if (lambda_io) *lambda_io = lambda;
            s.wi = wt.normalized();
            s.pdf = 1.f - F;
            s.weight = tint * spectral * (eta * eta);
            s.lobe = LOBE_TRANSMIT | LOBE_DELTA;
        }
        return s;
    }
    case BSDF_FORWARD:
    default: {
        // null BSDF: pass through unchanged (volume boundaries, forward.cuh)
        s.wi = -wo;
        s.pdf = 1.f;
        s.weight = Vec3(1.f);
        s.lobe = LOBE_TRANSMIT | LOBE_DELTA | LOBE_NULL;
        return s;
    }
    }
}

HD bool bsdf_is_delta(const BsdfParams& b) {
    return b.type == BSDF_SPECULAR || b.type == BSDF_TRANSLUCENT ||
           b.type == BSDF_DISPERSION || b.type == BSDF_FORWARD ||
           b.type == BSDF_PLASTIC_FORWARD;
}

} // namespace hippt
