// phase.h — phase functions: isotropic, Henyey-Greenstein, duo-HG mixture,
// Rayleigh, SGGX (isotropic placeholder, as in the reference).
//
// Capability parity: reference src/core/phase.cuh, src/volume/
// henyey_greenstein.cuh (analytic inverse CDF + duo-HG MIS mixture),
// rayleigh.cuh (exact inverse CDF via cbrt), sggx.cuh (placeholder).
#pragma once
#include "frame.h"
#include "sampling.h"
#include "rng.h"

namespace hippt {

enum PhaseType : int {
    PHASE_ISOTROPIC = 0,
    PHASE_HG,
    PHASE_DUO_HG,
    PHASE_RAYLEIGH,
    PHASE_SGGX,
    PHASE_NTYPES
};

struct alignas(16) PhaseParams {
    int32_t type;
    float g1, g2, wmix;   // HG asymmetry params; duo-HG mixture weight for g1
};

// HG phase value p(cos_theta), normalized over the sphere.
HD float hg_phase(float g, float cos_t) {
    // p(cos) = (1-g^2) / (4pi (1 + g^2 - 2 g cos)^(3/2)); forward-peaked for
    // g > 0, consistent with hg_sample_cos's inverse CDF
    float denom = 1.f + g * g - 2.f * g * cos_t;
    return (1.f / (4.f * PI)) * (1.f - g * g) / fmaxf(denom * sqrtf(denom), 1e-8f);
}

// cos_t = angle between incoming travel direction and outgoing direction.
HD float phase_eval(const PhaseParams& p, float cos_t) {
    switch (p.type) {
    case PHASE_HG: return hg_phase(p.g1, cos_t);
    case PHASE_DUO_HG: return p.wmix * hg_phase(p.g1, cos_t) + (1.f - p.wmix) * hg_phase(p.g2, cos_t);
    case PHASE_RAYLEIGH: return (3.f / (16.f * PI)) * (1.f + cos_t * cos_t);
    case PHASE_ISOTROPIC:
    case PHASE_SGGX:
    default: return 1.f / (4.f * PI);
    }
}

HD float hg_sample_cos(float g, float u) {
    if (fabsf(g) < 1e-3f) return 1.f - 2.f * u;
    float sq = (1.f - g * g) / (1.f - g + 2.f * g * u);
    return (1.f + g * g - sq * sq) / (2.f * g);
}

// Rayleigh exact inverse CDF (reference rayleigh.cuh:37-47).
HD float rayleigh_sample_cos(float u) {
    float x = 2.f * (2.f * u - 1.f);
    float s = sqrtf(x * x + 1.f);
    float A = cbrtf(x + s);
    return A - 1.f / A;
}

struct PhaseSampleRec {
    Vec3 wi;      // new travel direction
    float pdf;    // == phase value (perfect importance sampling except duo-HG)
    float weight; // phase/pdf (1 except duo-HG mixture)
};

// wo_travel = current propagation direction of the ray.
HD PhaseSampleRec phase_sample(const PhaseParams& p, const Vec3& wo_travel, Sampler& sp) {
    PhaseSampleRec r{};
    float cos_t;
    float u0 = sp.next1f();
    switch (p.type) {
    case PHASE_HG: cos_t = hg_sample_cos(p.g1, u0); break;
    case PHASE_DUO_HG: {
        float g = sp.next1f() < p.wmix ? p.g1 : p.g2;
        cos_t = hg_sample_cos(g, u0);
        break;
    }
    case PHASE_RAYLEIGH: cos_t = rayleigh_sample_cos(u0); break;
    default: cos_t = 1.f - 2.f * u0; break;
    }
    float sin_t = sqrtf(fmaxf(0.f, 1.f - cos_t * cos_t));
    float phi = TWO_PI * sp.next1f();
    Vec3 local{sin_t * cosf(phi), sin_t * sinf(phi), cos_t};
    r.wi = Frame::from_n(wo_travel).to_world(local);
    float val = phase_eval(p, cos_t);
    if (p.type == PHASE_DUO_HG) {
        // mixture importance sampling: pdf is the mixture pdf itself
        float pdf = p.wmix * hg_phase(p.g1, cos_t) + (1.f - p.wmix) * hg_phase(p.g2, cos_t);
        r.pdf = pdf;
        r.weight = pdf > 0.f ? val / pdf : 0.f;
    } else {
        r.pdf = val;
        r.weight = 1.f;
    }
    return r;
}

} // namespace hippt
