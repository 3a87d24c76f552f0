// v7 rollout: MFMA whole-episode rollout for the LINEAR flagship policy
// (K10+K11) — 16 members per workgroup.
//
// Why v7 beats v6 (2 members/block, all-v_dot2): v6 is LDS-bound — every
// dot2 reads 8 B of LDS for 4 FLOPs (~220 KB LDS traffic per block-step).
// The env dynamics (V·o, Uᵀh + D2ᵀa) have SHARED matrices across the
// population, i.e. real GEMMs once ≥16 members sit in one block:
//
//   h(16×16)    = obs(16×384) @ Vᵀ(384×16)          GEMM1, 12× mfma
//   o'(16×384)  = [h|act](16×64) @ [U;D2](64×384)    GEMM2, 48× mfma
//
// run on v_mfma_f32_16x16x32_bf16 with the STATIC B-operands (V, U, D2)
// pre-loaded into per-lane register fragments once per episode — zero LDS
// traffic for them in the steady state. The per-member policy GEMV
// (act = W·obsn, no shared operand — not MFMA-shaped) keeps per-member
// weights in REGISTERS (one 32-lane half-wave per member, 12 obs columns
// per lane, 17 accumulators) so each obsn element is read from LDS once
// per step instead of 17 times. Per-step LDS traffic drops ~3× and the
// member count per block rises 8×.
//
// Numerics contract: identical to v6 / rollout_eager (bf16 operands, fp32
// accumulate, quantize-then-normalize obs).
//
// Layout facts verified on-device (native_probes/mfma_probe.hip):
//   A(16×32): row = lane&15,  k = (lane>>4)*8 + i
//   B(32×16): col = lane&15,  k = (lane>>4)*8 + i
//   D(16×16): col = lane&15,  row = (lane>>4)*4 + reg

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "philox.h"

namespace ea {

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2_t;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float floatx4_t;

__device__ __forceinline__ __bf16 f2b7(float v) { return (__bf16)v; }
__device__ __forceinline__ float b2f7(__bf16 v) { return (float)v; }

// tanh via the hardware exp unit (v_exp_f32): ~1e-6 relative error —
// far below the bf16 quantization (2^-8) every value passes through.
// libm tanhf was 12 calls/lane/step of branchy code on the hot path.
__device__ __forceinline__ float tanh_fast(float x) {
    const float xc = fminf(fmaxf(x, -15.0f), 15.0f);
    const float e = __expf(2.0f * xc);
    return (e - 1.0f) / (e + 1.0f);
}

// One step of a 32-lane sum reduce on the VALU pipe via DPP modifiers —
// __shfl_down lowers to ds_bpermute_b32 (LDS pipe), and the policy
// phase's 17 accumulators × 5 levels were 85 LDS-pipe ops per lane per
// step, fighting the dot-product LDS reads. After row_shr 1/2/4/8 +
// row_bcast:15 the 32-lane totals sit in lanes 31 and 63.
template <int kCtrl>
__device__ __forceinline__ float dpp_add(float x) {
    const int moved = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, x), kCtrl, 0xf, 0xf, true);
    return x + __builtin_bit_cast(float, moved);
}

__device__ __forceinline__ float reduce32_dpp(float x) {
    x = dpp_add<0x111>(x);  // row_shr:1
    x = dpp_add<0x112>(x);  // row_shr:2
    x = dpp_add<0x114>(x);  // row_shr:4
    x = dpp_add<0x118>(x);  // row_shr:8
    x = dpp_add<0x142>(x);  // row_bcast:15 → lanes 31/63 hold the 32-lane sums
    return x;
}

struct RolloutV7Args {
    const float* params;     // [n][A*O + A]
    const float* env_blob;   // V[R][O] U_T[R][O] D2_T[A][O] c[O] wr[O] mean[O] std[O]
    float* fitness_out;      // [n]
    float* obs_stats_out;    // [2][O]
    int n_members;
    long member_offset;
    int obs_dim, act_dim, rank, steps;
    float alive_bonus, act_cost;
    unsigned long long init_seed;
};

// Compile-time geometry (all guards constant-fold; a runtime-guarded
// W-load was measured to demote w_frag to scratch memory): O padded to OP
// (multiple of 128 so 8 waves split OP/16 tiles evenly), R == 16 (one
// GEMM1 tile), A <= 31 (fits hact K=64).
template <int O, int A>
__global__ __launch_bounds__(512, 1) void rollout_v7_kernel(RolloutV7Args args) {
    constexpr int kMembers = 16;
    constexpr int OP = (O + 127) / 128 * 128;
    constexpr int A_MAX = A;
    constexpr int R = 16;
    constexpr int OPS = OP + 8;               // obs-row LDS stride: breaks the
                                              // 768 B ≡ 0 (mod 64 dwords) bank
                                              // aliasing of 16-row fragment reads
    constexpr int KS = 72;                    // hact/ud row stride (vs 64), same reason
    constexpr int kTiles = OP / 16;           // GEMM2 output tiles
    constexpr int kTilesPerWave = kTiles / 8; // = 3 at OP=384
    constexpr int kChunk = OP / 32;           // policy obs columns per lane (12)
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int base_member = blockIdx.x * kMembers;
    if (base_member >= args.n_members) return;
    const int live = min(kMembers, args.n_members - base_member);

    // ---- LDS ----
    extern __shared__ unsigned char lds7[];
    __bf16* obs_l = reinterpret_cast<__bf16*>(lds7);     // [16][OP] raw (quantized) obs
    __bf16* obsn_l = obs_l + kMembers * OPS;             // [16][OPS] normalized obs
    __bf16* hact_l = obsn_l + kMembers * OPS;            // [16][KS]: k<16 h, 16..16+A act, rest 0
    float* b_l = reinterpret_cast<float*>(hact_l + kMembers * KS);  // [16][A_MAX] bias
    float* c_l = b_l + kMembers * A_MAX;                 // [OP]
    float* wr_l = c_l + OP;                              // [OP]
    float* mean_l = wr_l + OP;                           // [OP]
    float* istd_l = mean_l + OP;                         // [OP]
    float* wave_fit = istd_l + OP;                       // [8][16] per-wave fitness partials
    float* actsq_l = wave_fit + 8 * 16;                  // [16]
    __bf16* v_l = reinterpret_cast<__bf16*>(actsq_l + 16);  // [16][OP] V (GEMM1 B-operand)
    __bf16* ud_l = v_l + kMembers * OPS;                 // [OP][KS] k-major [U;D2] (GEMM2 B)

    const long RO = (long)R * O, AO = (long)A * O;
    const float* eV = args.env_blob;
    const float* eU = eV + RO;
    const float* eD2 = eU + RO;
    const float* e_c = eD2 + AO;
    const float* e_wr = e_c + O;
    const float* e_mean = e_wr + O;
    const float* e_std = e_mean + O;

    // ---- stage vectors (pads: c=0, wr=0, mean=0, istd=0 → obsn pad = 0) ----
    for (int j = tid; j < OP; j += 512) {
        const bool in = j < O;
        c_l[j] = in ? e_c[j] : 0.0f;
        wr_l[j] = in ? e_wr[j] : 0.0f;
        mean_l[j] = in ? e_mean[j] : 0.0f;
        istd_l[j] = in ? 1.0f / e_std[j] : 0.0f;
    }
    for (int j = tid; j < kMembers * KS; j += 512) hact_l[j] = f2b7(0.0f);
    for (int j = tid; j < kMembers * A_MAX; j += 512) {
        const int m = j / A_MAX, a = j % A_MAX;
        b_l[j] = (m < live && a < A) ? args.params[(long)(base_member + m) * (AO + A) + AO + a] : 0.0f;
    }
    if (tid < kMembers) actsq_l[tid] = 0.0f;

    // ---- static B-operands ---------------------------------------------------
    // GEMM1 B (V) lives in LDS — keeping it in registers on every wave
    // (it is only used by wave 0) was what pushed the kernel over the
    // 256-VGPR budget and demoted w_frag to scratch.
    for (int j = tid; j < kMembers * OPS; j += 512) {
        const int r = j / OPS, o = j % OPS;
        v_l[j] = (r < R && o < O) ? f2b7(eV[(long)r * O + o]) : f2b7(0.0f);
    }
    // GEMM2 B ([U;D2]) in LDS, k-major [o][k] so a lane's 8-consecutive-k
    // fragment is one aligned 16 B read. (Register-resident B-fragments
    // for BOTH GEMMs pushed past the 256-VGPR/8-wave cap and demoted the
    // policy weights to scratch — LDS B costs ~49 KB/step of bandwidth
    // but keeps w_frag in registers, which dominates.)
    for (int j = tid; j < OP * KS; j += 512) {
        const int o = j / KS, k = j % KS;
        float v = 0.0f;
        if (o < O) {
            if (k < R) v = eU[(long)k * O + o];
            else if (k - R < A) v = eD2[(long)(k - R) * O + o];
        }
        ud_l[j] = f2b7(v);
    }

    // ---- per-member policy weights in registers -----------------------------
    // half-wave (32 lanes) per member: member = 2*wave + (lane>>5);
    // lane owns obsn columns [l32*kChunk, l32*kChunk + kChunk).
    const int l32 = lane & 31;
    const int my_member = 2 * wave + (lane >> 5);
    bf16x2_t w_frag[A_MAX][kChunk / 2];
    {
        const float* W = args.params + (long)(base_member + min(my_member, live - 1)) * ((long)A * O + A);
        const int cbase = l32 * kChunk;
#pragma unroll
        for (int a = 0; a < A_MAX; ++a) {
#pragma unroll
            for (int p = 0; p < kChunk / 2; ++p) {
                const int k0 = cbase + 2 * p;
                bf16x2_t w2;
                w2.x = (a < A && k0 < O) ? f2b7(W[(long)a * O + k0]) : f2b7(0.0f);
                w2.y = (a < A && k0 + 1 < O) ? f2b7(W[(long)a * O + k0 + 1]) : f2b7(0.0f);
                w_frag[a][p] = w2;
            }
        }
    }

    // ---- initial observations (philox stream per global member) -------------
    for (int j = tid; j < kMembers * OPS; j += 512) obs_l[j] = f2b7(0.0f);
    __syncthreads();
    {
        const int per_member4 = (O + 3) / 4;
        for (int idx = tid; idx < kMembers * per_member4; idx += 512) {
            const int m = idx / per_member4;
            const int j4 = idx % per_member4;
            if (m >= live) continue;
            float z[4];
            philox_normal4(args.init_seed, (uint32_t)(args.member_offset + base_member + m), (uint64_t)j4, z);
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                const int j = j4 * 4 + u;
                if (j < O) obs_l[m * OPS + j] = f2b7(0.1f * z[u]);
            }
        }
    }
    __syncthreads();
    for (int j = tid; j < kMembers * OPS; j += 512) {
        const int jo = j % OPS;
        obsn_l[j] = (jo < OP) ? f2b7((b2f7(obs_l[j]) - mean_l[jo]) * istd_l[jo]) : f2b7(0.0f);
    }
    __syncthreads();

    // ---- episode state in registers ----
    float fit_part[4] = {0.f, 0.f, 0.f, 0.f};  // member rows (lane>>4)*4+reg of GEMM2
    float actsq_total = 0.0f;                  // on l32==0 lanes: my_member's Σa²
    float stat_sum[kTilesPerWave] = {};        // per owned obs column
    float stat_sumsq[kTilesPerWave] = {};

    const int g2_row = lane & 15;              // A-frag row (member) for GEMM2/GEMM1
    const int g2_k0 = (lane >> 4) * 8;
    const int c_col = lane & 15;               // C-frag col
    const int c_row0 = (lane >> 4) * 4;        // C-frag first row (member)

    for (int t = 0; t < args.steps; ++t) {
        // ===== policy: act = clamp(W · obsn + b) — per-member half-waves =====
        {
            float acc[A_MAX];
#pragma unroll
            for (int a = 0; a < A_MAX; ++a) acc[a] = 0.0f;
            if (my_member < live) {
                const __bf16* on = obsn_l + my_member * OPS + l32 * kChunk;
#pragma unroll
                for (int p = 0; p < kChunk / 2; ++p) {
                    const bf16x2_t o2 = *reinterpret_cast<const bf16x2_t*>(on + 2 * p);
#pragma unroll
                    for (int a = 0; a < A_MAX; ++a) {
                        acc[a] = __builtin_amdgcn_fdot2_f32_bf16(w_frag[a][p], o2, acc[a], false);
                    }
                }
            }
#pragma unroll
            for (int a = 0; a < A_MAX; ++a) acc[a] = reduce32_dpp(acc[a]);
            if (l32 == 31 && my_member < live) {
                float sq = 0.0f;
#pragma unroll
                for (int a = 0; a < A_MAX; ++a) {
                    if (a >= A) break;
                    const float av = fminf(fmaxf(acc[a] + b_l[my_member * A_MAX + a], -1.0f), 1.0f);
                    hact_l[my_member * KS + 16 + a] = f2b7(av);
                    sq = fmaf(av, av, sq);
                }
                actsq_total += sq;
            }
        }
        // ===== GEMM1 (wave 0): h = obs @ Vᵀ =====
        if (wave == 0) {
            floatx4_t acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int s = 0; s < OP / 32; ++s) {
                const bf16x8_t a_frag = *reinterpret_cast<const bf16x8_t*>(obs_l + g2_row * OPS + s * 32 + g2_k0);
                const bf16x8_t b_frag = *reinterpret_cast<const bf16x8_t*>(v_l + c_col * OPS + s * 32 + g2_k0);
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0, 0, 0);
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                // D: col = h-index, row = member
                hact_l[(c_row0 + r) * KS + c_col] = f2b7(acc[r]);
            }
        }
        __syncthreads();

        // ===== GEMM2: o' = tanh(hact @ [U;D2] + c); fused epilogue =====
        {
            const bf16x8_t a0 = *reinterpret_cast<const bf16x8_t*>(hact_l + g2_row * KS + g2_k0);
            const bf16x8_t a1 = *reinterpret_cast<const bf16x8_t*>(hact_l + g2_row * KS + 32 + g2_k0);
            floatx4_t out[kTilesPerWave];
#pragma unroll
            for (int tw = 0; tw < kTilesPerWave; ++tw) {
                const int o = (wave * kTilesPerWave + tw) * 16 + c_col;
                const bf16x8_t b0 = *reinterpret_cast<const bf16x8_t*>(ud_l + o * KS + g2_k0);
                const bf16x8_t b1 = *reinterpret_cast<const bf16x8_t*>(ud_l + o * KS + 32 + g2_k0);
                floatx4_t acc = {0.f, 0.f, 0.f, 0.f};
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc, 0, 0, 0);
                out[tw] = acc;
            }
#pragma unroll
            for (int tw = 0; tw < kTilesPerWave; ++tw) {
                const int col = (wave * kTilesPerWave + tw) * 16 + c_col;
                const float cv = c_l[col];
                const float wrv = wr_l[col];
                const float mv = mean_l[col];
                const float iv = istd_l[col];
                const bool col_in = col < O;
                float ssum = 0.0f, ssq = 0.0f;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int m = c_row0 + r;
                    const float o_new = tanh_fast(out[tw][r] + cv);
                    fit_part[r] = fmaf(wrv, o_new, fit_part[r]);
                    if (m < live && col_in) {
                        ssum += o_new;
                        ssq = fmaf(o_new, o_new, ssq);
                    }
                    const __bf16 ob = f2b7(o_new);
                    obs_l[m * OPS + col] = ob;
                    obsn_l[m * OPS + col] = f2b7((b2f7(ob) - mv) * iv);
                }
                stat_sum[tw] += ssum;
                stat_sumsq[tw] += ssq;
            }
        }
        __syncthreads();
    }

    // ---- wrap-up ----
    // Deterministic fitness reduction (no float atomics — the kernel must
    // be bitwise run-to-run reproducible): (1) shuffle-reduce fit_part
    // across the 16 lanes of each row segment, (2) stage per-wave partials
    // in LDS, (3) one thread per member sums the 8 waves in fixed order.
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
#pragma unroll
        for (int r = 0; r < 4; ++r) fit_part[r] += __shfl_down(fit_part[r], off, 16);
    }
    if ((lane & 15) == 0) {
#pragma unroll
        for (int r = 0; r < 4; ++r) wave_fit[wave * 16 + c_row0 + r] = fit_part[r];
    }
    if (l32 == 31 && my_member < live) actsq_l[my_member] = actsq_total;
    __syncthreads();
    if (tid < live) {
        float total = 0.0f;
#pragma unroll
        for (int w = 0; w < 8; ++w) total += wave_fit[w * 16 + tid];
        args.fitness_out[base_member + tid] =
            total + args.alive_bonus * (float)args.steps - args.act_cost * actsq_l[tid] / (float)A;
    }
#pragma unroll
    for (int tw = 0; tw < kTilesPerWave; ++tw) {
        const int col = (wave * kTilesPerWave + tw) * 16 + c_col;
        if (col < O) {
            atomicAdd(&args.obs_stats_out[col], stat_sum[tw]);
            atomicAdd(&args.obs_stats_out[O + col], stat_sumsq[tw]);
        }
    }
}

void rollout_v7(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out, torch::Tensor fitness,
                int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus, double act_cost,
                int64_t init_seed, int64_t member_offset) {
    const int n = (int)params.size(0);
    const int O = (int)obs_dim, A = (int)act_dim, R = (int)rank;
    TORCH_CHECK(R == 16, "rollout v7 requires rank 16");
    TORCH_CHECK(A <= 17, "rollout v7 compiled for act_dim <= 17");
    TORCH_CHECK(O <= 384, "rollout v7 compiled for obs_dim <= 384");

    RolloutV7Args args;
    args.params = params.data_ptr<float>();
    args.env_blob = env_blob.data_ptr<float>();
    args.fitness_out = fitness.data_ptr<float>();
    args.obs_stats_out = obs_stats_out.data_ptr<float>();
    args.n_members = n;
    args.member_offset = (long)member_offset;
    args.obs_dim = O; args.act_dim = A; args.rank = R;
    args.steps = (int)steps;
    args.alive_bonus = (float)alive_bonus;
    args.act_cost = (float)act_cost;
    args.init_seed = (unsigned long long)init_seed;

    constexpr int O_T = 376, A_T = 17, OP = 384;
    TORCH_CHECK(O == O_T && A == A_T, "rollout v7 instantiated for the Humanoid geometry (obs 376, act 17)");
    const size_t lds = (size_t)(3 * 16 * (OP + 8) + 16 * 72 + OP * 72) * 2 + (size_t)(16 * A_T + 4 * OP + 8 * 16 + 16) * 4;
    static bool attr_set7 = false;
    if (!attr_set7) {
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&rollout_v7_kernel<O_T, A_T>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set7 = true;
    }
    auto stream = at::cuda::getCurrentCUDAStream();
    const int blocks = (n + 15) / 16;
    hipLaunchKernelGGL((rollout_v7_kernel<O_T, A_T>), dim3(blocks), dim3(512), lds, stream, args);
}

}  // namespace ea
