// fresnel.h — dielectric/conductor Fresnel terms + Snell refraction.
//
// Capability parity: reference src/bsdf/fresnel.cuh:39-119 (snell_refraction,
// is_total_reflection, fresnel_dielectric, RGB fresnel_conductor).
#pragma once
#include "vec.h"

namespace hippt {

// cos_i > 0 means the incident ray is on the outside (same side as normal).
HD float fresnel_dielectric(float cos_i, float eta_i, float eta_t) {
    cos_i = clampv(cos_i, -1.f, 1.f);
    if (cos_i < 0.f) { float t = eta_i; eta_i = eta_t; eta_t = t; cos_i = -cos_i; }
    float sin_i = sqrtf(fmaxf(0.f, 1.f - cos_i * cos_i));
    float sin_t = eta_i / eta_t * sin_i;
    if (sin_t >= 1.f) return 1.f;  // total internal reflection
    float cos_t = sqrtf(fmaxf(0.f, 1.f - sin_t * sin_t));
    float r_par = (eta_t * cos_i - eta_i * cos_t) / (eta_t * cos_i + eta_i * cos_t);
    float r_per = (eta_i * cos_i - eta_t * cos_t) / (eta_i * cos_i + eta_t * cos_t);
    return 0.5f * (r_par * r_par + r_per * r_per);
}

// Refract wi about n (unit, same side as wi). Returns false on TIR.
// eta = eta_i / eta_t (ratio of the side the ray comes from to the far side).
HD bool snell_refraction(const Vec3& wi, const Vec3& n, float eta, Vec3& wt) {
    float cos_i = wi.dot(n);
    float sin2_t = eta * eta * fmaxf(0.f, 1.f - cos_i * cos_i);
    if (sin2_t >= 1.f) return false;
    float cos_t = sqrtf(1.f - sin2_t);
    wt = (-wi) * eta + n * (eta * cos_i - cos_t);
    return true;
}

HD Vec3 reflect_dir(const Vec3& wo, const Vec3& n) {
    return n * (2.f * wo.dot(n)) - wo;
}

// Exact conductor Fresnel per channel (eta = n, k = extinction).
HD float fresnel_conductor_1(float cos_i, float eta, float k) {
    float c2 = cos_i * cos_i;
    float s2 = 1.f - c2;
    float e2 = eta * eta, k2 = k * k;
    float t0 = e2 - k2 - s2;
    float a2b2 = sqrtf(fmaxf(0.f, t0 * t0 + 4.f * e2 * k2));
    float t1 = a2b2 + c2;
    float a = sqrtf(fmaxf(0.f, 0.5f * (a2b2 + t0)));
    float t2 = 2.f * a * cos_i;
    float rs = (t1 - t2) / (t1 + t2);
    float t3 = c2 * a2b2 + s2 * s2;
    float t4 = t2 * s2;
    float rp = rs * (t3 - t4) / (t3 + t4);
    return 0.5f * (rp + rs);
}

HD Vec3 fresnel_conductor(float cos_i, const Vec3& eta, const Vec3& k) {
    cos_i = clampv(fabsf(cos_i), 0.f, 1.f);
    return {fresnel_conductor_1(cos_i, eta.x, k.x),
            fresnel_conductor_1(cos_i, eta.y, k.y),
            fresnel_conductor_1(cos_i, eta.z, k.z)};
}

// Average (hemispherical) diffuse Fresnel reflectance approximation used by
// the plastic BSDF interlayer (Egan/d'Eon style rational fit).
HD float fresnel_diffuse_avg(float eta) {
    // eta = ior ratio (>1 entering denser). Fit from d'Eon & Irving.
    if (eta >= 1.f) {
        return -1.4399f / (eta * eta) + 0.7099f / eta + 0.6681f + 0.0636f * eta;
    }
    float e2 = eta * eta, e3 = e2 * eta;
    return 0.919317f - 3.4793f * eta + 6.75335f * e2 - 7.80989f * e3 +
           4.98554f * e3 * eta - 1.36881f * e3 * e2;
}

} // namespace hippt
