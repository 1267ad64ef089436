// rng.h — per-path random number generator.
//
// Capability parity: reference src/core/sampler.cuh (TinySampler: 8-byte
// xorshift128+-style state).  Our Sampler keeps the same contract — 8 bytes of
// state, decorrelated per (pixel, spp, seed_offset), bit-identical streams on
// CPU and GPU — but uses our own construction: a splitmix32-seeded
// xorshift64* generator (public-domain constructions, implemented fresh).
#pragma once
#include "hd.h"

namespace hippt {

constexpr uint32_t SEED_SCALER = 11451u;  // per-frame seed stride

HD uint32_t splitmix32(uint32_t& s) {
    s += 0x9e3779b9u;
    uint32_t z = s;
    z = (z ^ (z >> 16)) * 0x85ebca6bu;
    z = (z ^ (z >> 13)) * 0xc2b2ae35u;
    return z ^ (z >> 16);
}

struct Sampler {
    uint64_t state;

    HD Sampler() : state(0x853c49e6748fea9bULL) {}
    // decorrelate with two rounds of splitmix over (index, seed)
    HD Sampler(uint32_t index, uint32_t seed) {
        uint32_t s = index * 0x9E3779B1u + seed * 0x7FEB352Du + 0x165667B1u;
        uint32_t lo = splitmix32(s);
        uint32_t hi = splitmix32(s);
        state = (uint64_t(hi) << 32) | lo;
        if (state == 0) state = 0x853c49e6748fea9bULL;
    }
    HD Sampler(uint64_t raw_state) : state(raw_state ? raw_state : 1u) {}

    HD uint32_t next_u32() {
        // xorshift64* step
        uint64_t x = state;
        x ^= x >> 12;
        x ^= x << 25;
        x ^= x >> 27;
        state = x;
        return uint32_t((x * 0x2545F4914F6CDD1DULL) >> 32);
    }
    // uniform in [0, 1)
    HD float next1f() { return float(next_u32() >> 8) * (1.f / 16777216.f); }
    HD Vec2 next2f() { float a = next1f(); float b = next1f(); return {a, b}; }
    HD Vec3 next3f() { float a = next1f(); float b = next1f(); float c = next1f(); return {a, b, c}; }
};

} // namespace hippt
