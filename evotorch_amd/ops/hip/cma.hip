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

// ---------------------------------------------------------------------------
// In-LDS Cholesky of one <=128x128 SPD diagonal panel (reference reaches
// this through torch.linalg.cholesky -> rocSOLVER, whose unblocked potf2
// runs a chain of tiny kernels that dominated the d=4096 CMA-ES profile —
// profiles/cma_phase_attribution_r2.log). ONE launch: stage the panel
// into LDS (odd row stride 133 dwords keeps column walks off the 64-bank
// alias), right-looking rank-1 factorization with the active column
// mirrored in a flat buffer (conflict-free broadcast reads in the
// trailing update), write back the lower triangle. The blocked driver
// (cmaes.py _blocked_cholesky) keeps rocBLAS trsm/gemm for the O(n³)
// mass and calls this for every diagonal panel, so no rocSOLVER kernel
// remains in the factorization.

constexpr int kPotrfMax = 128;
constexpr int kPotrfPad = kPotrfMax + 5;  // odd dword stride: 133 mod 64 = 5 (measured 0.51 conflict cycles/LDS instr at +4)

constexpr int kPotrfBw = 8;        // micro-panel width (rank of each trailing update)
constexpr int kPotrfPPad = kPotrfBw + 1;  // stripe mirror row stride (conflict-free column reads)

__global__ __launch_bounds__(256) void potrf_panel_kernel(float* __restrict__ A, long lda, int n,
                                                          int* __restrict__ info) {
    // Rank-8 right-looking factorization: a column-at-a-time scheme costs
    // 2 block barriers per column (measured 485 us for n=128 — a pure
    // latency chain). Here wave 0 factors an 8-column stripe
    // wave-synchronously (lockstep lanes, in-order LDS, scheduling
    // fences), then all 8 waves apply ONE rank-8 trailing update: 2
    // barriers per 8 columns. The stripe is mirrored into a compact
    // (n x 9)-stride buffer so the trailing update's column reads are
    // bank-conflict-free.
    extern __shared__ float T[];  // kPotrfMax*kPotrfPad | stripe mirror kPotrfMax*kPotrfPPad
    float* P = T + kPotrfMax * kPotrfPad;
    const int tid = threadIdx.x;
    for (int e = tid; e < n * n; e += 256) {
        const int r = e / n, c = e % n;
        T[r * kPotrfPad + c] = A[(long)r * lda + c];
    }
    __syncthreads();
    for (int jb = 0; jb < n; jb += kPotrfBw) {
        const int bw = min(kPotrfBw, n - jb);
        if (tid < 64) {
            for (int c = 0; c < bw; ++c) {
                const int j = jb + c;
                const float d = T[j * kPotrfPad + j];
                const float s = sqrtf(d);
                const float inv = 1.0f / s;
                if (tid == 0) {
                    if (!(d > 0.0f)) atomicCAS(info, 0, j + 1);
                    T[j * kPotrfPad + j] = s;
                    P[j * kPotrfPPad + c] = s;
                }
                for (int i = j + 1 + tid; i < n; i += 64) {
                    const float v = T[i * kPotrfPad + j] * inv;
                    T[i * kPotrfPad + j] = v;
                    P[i * kPotrfPPad + c] = v;
                }
                // lanes read other lanes' stripe stores next: keep the
                // LDS ops in program order (wave lockstep + in-order LDS
                // make that sufficient)
                __builtin_amdgcn_wave_barrier();
                for (int i = j + 1 + tid; i < n; i += 64) {
                    const float li = P[i * kPotrfPPad + c];
                    for (int cc = c + 1; cc < bw; ++cc)
                        T[i * kPotrfPad + jb + cc] =
                            fmaf(-li, P[(jb + cc) * kPotrfPPad + c], T[i * kPotrfPad + jb + cc]);
                }
                __builtin_amdgcn_wave_barrier();
            }
        }
        __syncthreads();
        const int j1 = jb + bw;
        const int m = n - j1;
        if (bw == kPotrfBw) {
            for (int e = tid; e < m * m; e += 256) {
                const int r = j1 + e / m, c = j1 + e % m;
                if (r < c) continue;
                float acc = T[r * kPotrfPad + c];
#pragma unroll
                for (int k = 0; k < kPotrfBw; ++k)
                    acc = fmaf(-P[r * kPotrfPPad + k], P[c * kPotrfPPad + k], acc);
                T[r * kPotrfPad + c] = acc;
            }
        } else {
            for (int e = tid; e < m * m; e += 256) {
                const int r = j1 + e / m, c = j1 + e % m;
                if (r < c) continue;
                float acc = T[r * kPotrfPad + c];
                for (int k = 0; k < bw; ++k)
                    acc = fmaf(-P[r * kPotrfPPad + k], P[c * kPotrfPPad + k], acc);
                T[r * kPotrfPad + c] = acc;
            }
        }
        __syncthreads();
    }
    for (int e = tid; e < n * n; e += 256) {
        const int r = e / n, c = e % n;
        if (r >= c) A[(long)r * lda + c] = T[r * kPotrfPad + c];
    }
}

void potrf_tile(torch::Tensor A, torch::Tensor info) {
    CHECK_GPU_C(A);
    TORCH_CHECK(A.dim() == 2 && A.size(0) == A.size(1), "A must be square");
    TORCH_CHECK(A.size(0) <= kPotrfMax, "panel larger than ", kPotrfMax);
    TORCH_CHECK(A.stride(1) == 1, "A rows must be contiguous");
    TORCH_CHECK(A.scalar_type() == at::ScalarType::Float, "A must be fp32");
    TORCH_CHECK(info.is_cuda() && info.scalar_type() == at::ScalarType::Int && info.numel() >= 1,
                "info must be a device int32 scalar");
    const int n = (int)A.size(0);
    const size_t lds = (size_t)(kPotrfMax * kPotrfPad + kPotrfMax * kPotrfPPad) * sizeof(float);
    static bool attr_set = false;
    if (!attr_set) {
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&potrf_panel_kernel),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set = true;
    }
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(potrf_panel_kernel, dim3(1), dim3(256), lds, stream, A.data_ptr<float>(),
                       (long)A.stride(0), n, info.data_ptr<int>());
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
