// Counter-based RNG: philox4x32-10 (Salmon et al., SC'11) + Box-Muller.
// Counter-based generation is what makes population sampling reproducible
// when the population is sharded across ranks: rank r generating rows
// [r*n, (r+1)*n) uses the same (seed, element-index) -> value mapping as a
// single process generating the whole population (SURVEY.md §7 "RNG
// discipline").
#pragma once
#include <hip/hip_runtime.h>

namespace ea {

struct uint4_philox {
    uint32_t x, y, z, w;
};

__device__ __forceinline__ uint4_philox philox4x32(uint32_t c0, uint32_t c1, uint32_t c2, uint32_t c3,
                                                   uint32_t k0, uint32_t k1) {
    const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
    for (int round = 0; round < 10; ++round) {
        uint32_t h0 = __umulhi(M0, c0), l0 = M0 * c0;
        uint32_t h1 = __umulhi(M1, c2), l1 = M1 * c2;
        uint32_t n0 = h1 ^ c1 ^ k0;
        uint32_t n1 = l1;
        uint32_t n2 = h0 ^ c3 ^ k1;
        uint32_t n3 = l0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += W0; k1 += W1;
    }
    return {c0, c1, c2, c3};
}

// uint32 -> uniform in (0, 1]  (never 0, so log() is safe)
__device__ __forceinline__ float u32_to_uniform(uint32_t v) {
    return (static_cast<float>(v >> 8) + 1.0f) * (1.0f / 16777216.0f);
}

// 4 uniform u32 -> 4 standard normals via two Box-Muller transforms.
// Uses the fast-math intrinsics (__logf / __sincosf, ~1e-6 relative
// error): population sampling is write-bound and statistical; the CPU
// philox reference matches within the test tolerances.
__device__ __forceinline__ void box_muller4(uint4_philox r, float out[4]) {
    float u0 = u32_to_uniform(r.x), u1 = u32_to_uniform(r.y);
    float u2 = u32_to_uniform(r.z), u3 = u32_to_uniform(r.w);
    float r0 = sqrtf(-2.0f * __logf(u0));
    float r1 = sqrtf(-2.0f * __logf(u2));
    float s0, c0, s1, c1;
    __sincosf(6.2831853071795864f * u1, &s0, &c0);
    __sincosf(6.2831853071795864f * u3, &s1, &c1);
    out[0] = r0 * c0;
    out[1] = r0 * s0;
    out[2] = r1 * c1;
    out[3] = r1 * s1;
}

// 4 standard normals for global element block `idx4` (covers elements
// [4*idx4, 4*idx4+4)) of the stream identified by `seed` / `stream_id`.
__device__ __forceinline__ void philox_normal4(uint64_t seed, uint32_t stream_id, uint64_t idx4, float out[4]) {
    uint4_philox r = philox4x32(static_cast<uint32_t>(idx4), static_cast<uint32_t>(idx4 >> 32), stream_id, 0u,
                                static_cast<uint32_t>(seed), static_cast<uint32_t>(seed >> 32));
    box_muller4(r, out);
}

}  // namespace ea
