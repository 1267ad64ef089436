// spectrum.h — wavelength <-> RGB conversion + blackbody emission.
//
// Capability parity: reference src/core/xyz.cuh + impl/xyz.cu (CIE 1931
// tables as 1D textures, used by the dispersion BSDF) and the blackbody
// emission table (vol_grid.cu:97-102).  Instead of shipping 471-sample
// tables, we use the multi-lobe Gaussian analytic fits of the CIE 1931
// color-matching functions (Wyman, Sloan & Shirley, JCGT 2013) — closed form,
// register-cheap, accurate to ~1% which is far below path-tracing noise.
#pragma once
#include "vec.h"

namespace hippt {

HD float cie_gauss(float x, float alpha, float mu, float s1, float s2) {
    float t = (x - mu) * (x < mu ? 1.f / s1 : 1.f / s2);
    return alpha * expf(-0.5f * t * t);
}

// CIE 1931 standard observer color matching functions, lambda in nm.
HD float cie_x(float l) {
    return cie_gauss(l, 1.056f, 599.8f, 37.9f, 31.0f) +
           cie_gauss(l, 0.362f, 442.0f, 16.0f, 26.7f) +
           cie_gauss(l, -0.065f, 501.1f, 20.4f, 26.2f);
}
HD float cie_y(float l) {
    return cie_gauss(l, 0.821f, 568.8f, 46.9f, 40.5f) +
           cie_gauss(l, 0.286f, 530.9f, 16.3f, 31.1f);
}
HD float cie_z(float l) {
    return cie_gauss(l, 1.217f, 437.0f, 11.8f, 36.0f) +
           cie_gauss(l, 0.681f, 459.0f, 26.0f, 13.8f);
}

// Linear sRGB from XYZ (reference xyz.cuh:48-57 XYZ_to_sRGB).
HD Vec3 xyz_to_srgb(const Vec3& c) {
    return {fmaf(3.2404542f, c.x, fmaf(-1.5371385f, c.y, -0.4985314f * c.z)),
            fmaf(-0.9692660f, c.x, fmaf(1.8760108f, c.y, 0.0415560f * c.z)),
            fmaf(0.0556434f, c.x, fmaf(-0.2040259f, c.y, 1.0572252f * c.z))};
}

// RGB weight of a single sampled wavelength (uniform in [LAMBDA_MIN, LAMBDA_MAX]).
constexpr float LAMBDA_MIN = 360.f, LAMBDA_MAX = 830.f;
HD Vec3 wavelength_to_rgb(float lambda_nm) {
    Vec3 xyz{cie_x(lambda_nm), cie_y(lambda_nm), cie_z(lambda_nm)};
    // normalize so a flat spectrum integrates to ~white.  Do NOT clamp the
    // negative out-of-gamut sRGB lobes: the single-wavelength estimator is
    // linear and relies on them canceling in expectation — clamping per
    // sample was measured to ADD ~30% red energy in the dispersion furnace.
    return xyz_to_srgb(xyz * ((LAMBDA_MAX - LAMBDA_MIN) / 106.857f));
}

// Planck blackbody radiance -> linear sRGB (normalized to luminance ~1 at the
// given temperature scale); T in Kelvin.  Replaces the reference's
// blackbody.bin 1D texture with a closed-form 16-sample quadrature.
HD Vec3 blackbody_rgb(float T) {
    if (T <= 0.f) return Vec3(0.f);
    Vec3 xyz(0.f);
    const float c2 = 1.4388e7f;  // nm*K  (hc/kB)
    for (int i = 0; i < 16; ++i) {
        float l = 380.f + (i + 0.5f) * (400.f / 16.f);  // 380..780nm
        // relative Planck spectrum (1e27 scaling folded away by normalization)
        float x = c2 / (l * T);
        float p = 1.f / (l * l * l * l * l * (expf(x) - 1.f));
        xyz += Vec3(cie_x(l), cie_y(l), cie_z(l)) * p;
    }
    float norm = fmaxf(xyz.y, 1e-30f);
    Vec3 rgb = xyz_to_srgb(xyz / norm);
    return rgb.maxv(Vec3(0.f));
}

} // namespace hippt
