// K5: fused CMA-ES covariance update (reference
// cmaes.py:519-565):
//
//   C' = scale·C + c1·pc·pcᵀ + cμ·Σ_k w_k·y_k·y_kᵀ
//
// Design (MI355X): at CMA-ES shapes the update is MEMORY-bound, not
// matrix-core-bound — the K dimension is the population (λ ≲ 10²) while
// C is d×d (d up to ~8k, 32 MB+ fp32), and CDNA4 has no fp32-input MFMA
// (cdna_hip_programming.md §3; bf16-quantizing a covariance accumulated
// over thousands of generations is numerically unacceptable). The torch
// expression chain costs ~7 d² passes (GEMM out, outer-product out,
// scale+add chain, explicit symmetrization). This kernel does ONE pass:
// each workgroup produces a 64×64 tile of the UPPER triangle with the
// w-scaled Y panels staged through LDS, applies all three terms in the
// epilogue, and writes the tile AND its mirror — exact symmetry by
// construction, half the FLOPs, ~4× less HBM traffic.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace ea {

#define CHECK_GPU_C(x) TORCH_CHECK(x.is_cuda(), #x " must be a ROCm tensor")

constexpr int kTile = 64;
constexpr int kChunkK = 32;  // Y rows staged per LDS round

__global__ __launch_bounds__(256) void cma_update_c_kernel(
    float* __restrict__ C, const float* __restrict__ Y, const float* __restrict__ w,
    const float* __restrict__ pc, const float* __restrict__ hs_f, const float* __restrict__ wsum,
    int d, int lam, float c1, float cmu, float cc) {
    // scale computed from DEVICE scalars (hs stall flag, Σw) — reading
    // them host-side would force a sync every generation
    const float delta_hs = (1.0f - *hs_f) * cc * (2.0f - cc);
    const float scale = 1.0f + c1 * delta_hs - c1 - cmu * (*wsum);
    // upper-triangular tile decode: blockIdx.x -> (ti, tj), ti <= tj
    const int ntiles = (d + kTile - 1) / kTile;
    int b = blockIdx.x;
    int ti = 0;
    // row ti owns (ntiles - ti) tiles; walk rows (ntiles <= 128 for d<=8k)
    while (b >= ntiles - ti) {
        b -= ntiles - ti;
        ++ti;
    }
    const int tj = ti + b;
    const int i0 = ti * kTile, j0 = tj * kTile;

    __shared__ float Yi[kChunkK][kTile];  // w-scaled rows, i-panel
    __shared__ float Yj[kChunkK][kTile];  // raw rows, j-panel

    const int tid = threadIdx.x;
    float acc[16];
#pragma unroll
    for (int q = 0; q < 16; ++q) acc[q] = 0.0f;

    for (int k0 = 0; k0 < lam; k0 += kChunkK) {
        const int kc = min(kChunkK, lam - k0);
        for (int e = tid; e < kc * kTile; e += 256) {
            const int kk = e / kTile, col = e % kTile;
            const int gi = i0 + col, gj = j0 + col;
            const float wk = w[k0 + kk];
            Yi[kk][col] = (gi < d) ? wk * Y[(long)(k0 + kk) * d + gi] : 0.0f;
            Yj[kk][col] = (gj < d) ? Y[(long)(k0 + kk) * d + gj] : 0.0f;
        }
        __syncthreads();
#pragma unroll 4
        for (int kk = 0; kk < kChunkK; ++kk) {
            if (kk >= kc) break;
#pragma unroll
            for (int q = 0; q < 16; ++q) {
                const int e = tid + 256 * q;
                const int r = e >> 6, c = e & 63;
                acc[q] = fmaf(Yi[kk][r], Yj[kk][c], acc[q]);
            }
        }
        __syncthreads();
    }

#pragma unroll
    for (int q = 0; q < 16; ++q) {
        const int e = tid + 256 * q;
        const int r = e >> 6, c = e & 63;
        const int gi = i0 + r, gj = j0 + c;
        if (gi >= d || gj >= d) continue;
        if (ti == tj && gi > gj) continue;  // diagonal tile: upper half only
        const float cnew = fmaf(scale, C[(long)gi * d + gj], fmaf(c1 * pc[gi], pc[gj], cmu * acc[q]));
        C[(long)gi * d + gj] = cnew;
        if (gi != gj) C[(long)gj * d + gi] = cnew;
    }
}

void cma_update_c(torch::Tensor C, torch::Tensor Y, torch::Tensor w, torch::Tensor pc, torch::Tensor hs_f,
                  torch::Tensor wsum, double c1, double cmu, double cc) {
    CHECK_GPU_C(C);
    TORCH_CHECK(C.is_contiguous() && C.dim() == 2 && C.size(0) == C.size(1), "C must be contiguous square");
    TORCH_CHECK(C.scalar_type() == at::ScalarType::Float, "C must be fp32");
    TORCH_CHECK(Y.is_contiguous() && Y.scalar_type() == at::ScalarType::Float, "Y must be contiguous fp32");
    const int d = (int)C.size(0);
    const int lam = (int)Y.size(0);
    TORCH_CHECK(Y.size(1) == d && w.numel() == lam && pc.numel() == d, "shape mismatch");
    const int ntiles = (d + kTile - 1) / kTile;
    const int nblocks = ntiles * (ntiles + 1) / 2;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(cma_update_c_kernel, dim3(nblocks), dim3(256), 0, stream, C.data_ptr<float>(),
                       Y.data_ptr<float>(), w.data_ptr<float>(), pc.data_ptr<float>(), hs_f.data_ptr<float>(),
                       wsum.data_ptr<float>(), d, lam, (float)c1, (float)cmu, (float)cc);
}

}  // namespace ea
