// Wave- and block-level reduction helpers for CDNA4 (64-wide wavefronts).
#pragma once
#include <hip/hip_runtime.h>

namespace ea {

constexpr int kWaveSize = 64;  // CDNA wavefront — NOT 32

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int offset = kWaveSize / 2; offset > 0; offset >>= 1) {
        v += __shfl_down(v, offset, kWaveSize);
    }
    return v;
}

// Block reduction; `scratch` must hold >= blockDim.x/64 floats.
// Result is valid on thread 0 (and broadcast to all if broadcast=true).
template <bool kBroadcast = false>
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
    const int lane = threadIdx.x & (kWaveSize - 1);
    const int wave = threadIdx.x / kWaveSize;
    const int nwaves = (blockDim.x + kWaveSize - 1) / kWaveSize;
    v = wave_reduce_sum(v);
    if (lane == 0) scratch[wave] = v;
    __syncthreads();
    float total = 0.0f;
    if (wave == 0) {
        float x = (lane < nwaves) ? scratch[lane] : 0.0f;
        total = wave_reduce_sum(x);
        if (kBroadcast && lane == 0) scratch[0] = total;
    }
    if (kBroadcast) {
        __syncthreads();
        total = scratch[0];
    }
    return total;
}

}  // namespace ea
