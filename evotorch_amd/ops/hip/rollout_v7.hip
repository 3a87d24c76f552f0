// v7 rollout: MFMA whole-episode rollout for the LINEAR flagship policy
// (K10+K11).
//
// Why v7 beats v6 (2 members/block, all-v_dot2): v6 is LDS-bound — every
// dot2 reads 8 B of LDS for 4 FLOPs (~220 KB LDS traffic per block-step).
// The env dynamics (V·o, Uᵀh + D2ᵀa) have SHARED matrices across the
// population, i.e. real GEMMs once many members sit in one block:
//
//   h(M×16)    = obs(M×384) @ Vᵀ(384×16)          GEMM1 on mfma
//   o'(M×384)  = [h|act](M×64) @ [U;D2](64×384)    GEMM2 on mfma
//
// run on v_mfma_f32_16x16x32_bf16. The per-member policy GEMV
// (act = W·obsn, no shared operand — not MFMA-shaped) keeps per-member
// weights in REGISTERS (one 32-lane half-wave per member, 12 obs columns
// per lane, 17 accumulators, DPP-reduced on the VALU pipe).
//
// Occupancy (the v7.2→v7.3 lesson, measured via SQ_WAIT_ANY:SQ_BUSY ≈ 5:1):
// a single 8-wave block per CU exposes every barrier and LDS-latency chain
// of the 1000-step loop — there is nothing else to run. v7.3 uses 4-wave
// (256-thread) blocks of 8 members with ≤ 80 KB LDS and ≤ 256 VGPRs, so
// TWO independent blocks share each CU and cover each other's stalls; the
// GEMM2 B-operand ([U;D2]) lives in register fragments (48 VGPRs), the
// GEMM1 B-operand (V) in LDS.
//
// LDS row strides are padded (OPS = OP+8, KS = 72) — the natural 768 B /
// 128 B strides are ≡ 0 (mod 64 dwords), serializing 16-row fragment
// reads up to 16-way (measured: SQ_LDS_BANK_CONFLICT ≈ 2× SQ_BUSY).
//
// Numerics contract: identical to v6 / rollout_eager (bf16 operands, fp32
// accumulate, quantize-then-normalize obs); tanh via the hardware exp
// unit (~1e-6 rel. error, far below the bf16 quantization of every
// operand).
//
// Layout facts verified on-device (native_probes/mfma_probe.hip):
//   A(16×32): row = lane&15,  k = (lane>>4)*8 + i
//   B(32×16): col = lane&15,  k = (lane>>4)*8 + i
//   D(16×16): col = lane&15,  row = (lane>>4)*4 + reg

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdlib>

#include "philox.h"

namespace ea {

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2_t;
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
    // v_rcp_f32 (~1 ulp) instead of an IEEE divide: the result is bf16-
    // quantized immediately, so the 2^-23-level rcp error is invisible
    return (e - 1.0f) * __builtin_amdgcn_rcpf(e + 1.0f);
}

// 32-lane sum reduce on the VALU pipe via DPP modifiers — __shfl_down
// lowers to ds_bpermute_b32 (LDS pipe), and the policy phase's 17
// accumulators × 5 levels were 85 LDS-pipe ops per lane per step,
// fighting the dot-product LDS reads. After row_shr 1/2/4/8 +
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
    float* obs_stats_out;    // [blocks][2][O] per-block partials (deterministic sum in the launcher)
    int n_members;
    long member_offset;
    int obs_dim, act_dim, rank, steps;
    float alive_bonus, act_cost;
    unsigned long long init_seed;
    const unsigned long long* seed_ptr;  // device episode seed (hipGraph-safe); overrides init_seed
    int skip_mask;  // perf probe only (EVOTORCH_AMD_V7_SKIP): 1=policy 2=GEMM1 4=GEMM2
};

// Compile-time geometry (all guards constant-fold; a runtime-guarded
// W-load was measured to demote w_frag to scratch memory). kWaves ∈ {4, 8}:
// members per block = 2·kWaves.
template <int O, int A, int kWaves>
__global__ __launch_bounds__(64 * kWaves, 1) void rollout_v7_kernel(RolloutV7Args args) {
    constexpr int kThreads = 64 * kWaves;
    constexpr int kM = 2 * kWaves;            // members per block
    constexpr int OP = (O + 127) / 128 * 128;
    constexpr int OPS = OP + 8;               // padded obs-row stride (bank decorrelation)
    constexpr int KS = 72;                    // padded hact row stride (vs 64)
    constexpr int A_MAX = A;
    constexpr int R = 16;
    constexpr int kTiles = OP / 16;           // GEMM2 output tiles
    constexpr int kTilesPerWave = kTiles / kWaves;
    constexpr int kChunk = OP / 32;           // policy obs columns per lane (12)
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int base_member = blockIdx.x * kM;
    if (base_member >= args.n_members) return;
    const int live = min(kM, args.n_members - base_member);

    // ---- LDS ----
    extern __shared__ unsigned char lds7[];
    __bf16* obs_l = reinterpret_cast<__bf16*>(lds7);     // [kM][OPS] raw (quantized) obs
    __bf16* obsn_l = obs_l + kM * OPS;                   // [kM][OPS] normalized obs
    __bf16* hact_l = obsn_l + kM * OPS;                  // [16][KS]: k<16 h, k=16..31 act[0..16), slot 32 = act[16]
    __bf16* v_l = hact_l + 16 * KS;                      // [16][OPS] V (GEMM1 B-operand)
    float* b_l = reinterpret_cast<float*>(v_l + 16 * OPS);  // [kM][A_MAX] bias
    float* c_l = b_l + kM * A_MAX;                       // [OP]
    float* wr_l = c_l + OP;                              // [OP]
    float* mean_l = wr_l + OP;                           // [OP]
    float* istd_l = mean_l + OP;                         // [OP]
    float* wave_fit = istd_l + OP;                       // [kWaves][16] fitness partials
    float* actsq_l = wave_fit + kWaves * 16;             // [kM]
    float* d2last_l = actsq_l + kM;                      // [OP] D2_T row A-1 (rank-1 epilogue term)
    __bf16* ud_l = reinterpret_cast<__bf16*>(d2last_l + OP);  // [OP][40] k-major GEMM2 B

    const long RO = (long)R * O, AO = (long)A * O;
    const float* eV = args.env_blob;
    const float* eU = eV + RO;
    const float* eD2 = eU + RO;
    const float* e_c = eD2 + AO;
    const float* e_wr = e_c + O;
    const float* e_mean = e_wr + O;
    const float* e_std = e_mean + O;

    // ---- stage vectors (pads: c=0, wr=0, mean=0, istd=0 → obsn pad = 0) ----
    for (int j = tid; j < OP; j += kThreads) {
        const bool in = j < O;
        c_l[j] = in ? e_c[j] : 0.0f;
        wr_l[j] = in ? e_wr[j] : 0.0f;
        mean_l[j] = in ? e_mean[j] : 0.0f;
        istd_l[j] = in ? 1.0f / e_std[j] : 0.0f;
        // act[A-1]'s D2 row is applied as a rank-1 epilogue term (bf16-
        // quantized like the mfma operands) so the GEMM K stays 32
        d2last_l[j] = in ? b2f7(f2b7(eD2[(long)(A - 1) * O + j])) : 0.0f;
    }
    for (int j = tid; j < 16 * KS; j += kThreads) hact_l[j] = f2b7(0.0f);
    for (int j = tid; j < kM * A_MAX; j += kThreads) {
        const int m = j / A_MAX, a = j % A_MAX;
        b_l[j] = (m < live && a < A) ? args.params[(long)(base_member + m) * (AO + A) + AO + a] : 0.0f;
    }
    if (tid < kM) actsq_l[tid] = 0.0f;

    // GEMM1 B (V) in LDS (only wave 0 reads it — registers on every wave
    // would blow the VGPR budget and demote w_frag to scratch).
    for (int j = tid; j < 16 * OPS; j += kThreads) {
        const int r = j / OPS, o = j % OPS;
        v_l[j] = (r < R && o < O) ? f2b7(eV[(long)r * O + o]) : f2b7(0.0f);
    }

    const int g2_row = lane & 15;              // A-frag row (member) for GEMM2/GEMM1
    const int g2_k0 = (lane >> 4) * 8;
    const int c_col = lane & 15;               // C-frag col
    const int c_row0 = (lane >> 4) * 4;        // C-frag first row (member)

    // GEMM2 B ([U;D2 rows 0..15]): K = 32 exactly (R=16 h-rows + 16 action
    // rows; action A-1 is the rank-1 epilogue term), k-major LDS with
    // padded stride 40. Register fragments would push the 4-wave variant
    // past 256 VGPRs and forfeit its 2-blocks/CU occupancy.
    constexpr int KU = 40;  // ud_l row stride: 32 + 8 pad (bank decorrelation)
    for (int j = tid; j < OP * KU; j += kThreads) {
        const int o = j / KU, k = j % KU;
        float v = 0.0f;
        if (o < O && k < 32) {
            if (k < R) v = eU[(long)k * O + o];
            else if (k - R < A - 1) v = eD2[(long)(k - R) * O + o];
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
    for (int j = tid; j < kM * OPS; j += kThreads) obs_l[j] = f2b7(0.0f);
    __syncthreads();
    {
        const int per_member4 = (O + 3) / 4;
        for (int idx = tid; idx < kM * per_member4; idx += kThreads) {
            const int m = idx / per_member4;
            const int j4 = idx % per_member4;
            if (m >= live) continue;
            float z[4];
            const unsigned long long iseed = args.seed_ptr ? *args.seed_ptr : args.init_seed;
            philox_normal4(iseed, (uint32_t)(args.member_offset + base_member + m), (uint64_t)j4, z);
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                const int j = j4 * 4 + u;
                if (j < O) obs_l[m * OPS + j] = f2b7(0.1f * z[u]);
            }
        }
    }
    __syncthreads();
    for (int j = tid; j < kM * OPS; j += kThreads) {
        const int jo = j % OPS;
        obsn_l[j] = (jo < OP) ? f2b7((b2f7(obs_l[j]) - mean_l[jo]) * istd_l[jo]) : f2b7(0.0f);
    }
    __syncthreads();

    // ---- episode state in registers ----
    float fit_part[4] = {0.f, 0.f, 0.f, 0.f};  // member rows (lane>>4)*4+reg of GEMM2
    float actsq_total = 0.0f;                  // on l32==31 lanes: my_member's Σa²
    float stat_sum[kTilesPerWave] = {};        // per owned obs column
    float stat_sumsq[kTilesPerWave] = {};

    // A-frag rows beyond the stored kM member rows alias row 0 (safe reads;
    // their outputs are masked in every epilogue).
    const int a_row = (g2_row < kM) ? g2_row : 0;

    for (int t = 0; t < args.steps; ++t) {
        // ===== GEMM1 (wave 0): h = obs @ Vᵀ — issued before the policy VALU
        // work so the MFMA chain executes on the MAI pipe underneath it =====
        floatx4_t h_acc = {0.f, 0.f, 0.f, 0.f};
        if (wave == 0 && !(args.skip_mask & 2)) {
            // two accumulator chains: a single accumulator serializes 12
            // dependent MFMAs (~32 cycles each) on wave 0, which is the
            // barrier straggler (it also runs its policy share)
            floatx4_t h_acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int s = 0; s < OP / 32; s += 2) {
                const bf16x8_t a0 = *reinterpret_cast<const bf16x8_t*>(obs_l + a_row * OPS + s * 32 + g2_k0);
                const bf16x8_t b0 = *reinterpret_cast<const bf16x8_t*>(v_l + c_col * OPS + s * 32 + g2_k0);
                h_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, h_acc, 0, 0, 0);
                const bf16x8_t a1 = *reinterpret_cast<const bf16x8_t*>(obs_l + a_row * OPS + (s + 1) * 32 + g2_k0);
                const bf16x8_t b1 = *reinterpret_cast<const bf16x8_t*>(v_l + c_col * OPS + (s + 1) * 32 + g2_k0);
                h_acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, h_acc1, 0, 0, 0);
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) h_acc[r] += h_acc1[r];
        }
        // ===== policy: act = clamp(W · obsn + b) — per-member half-waves =====
        if (!(args.skip_mask & 1)) {
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
                    const float av = fminf(fmaxf(acc[a] + b_l[my_member * A_MAX + a], -1.0f), 1.0f);
                    hact_l[my_member * KS + ((a < 16) ? (16 + a) : 32)] = f2b7(av);
                    sq = fmaf(av, av, sq);
                }
                actsq_total += sq;
            }
        }
        if (wave == 0) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                // D: col = h-index, row = member
                const int m = c_row0 + r;
                if (m < kM) hact_l[m * KS + c_col] = f2b7(h_acc[r]);
            }
        }
        if (!(args.skip_mask & 8)) __syncthreads();

        // ===== GEMM2: o' = tanh(hact @ [U;D2] + act16·d2last + c) =====
        if (!(args.skip_mask & 4)) {
            const bf16x8_t a0 = *reinterpret_cast<const bf16x8_t*>(hact_l + g2_row * KS + g2_k0);
            // per-member act[A-1] for the rank-1 term, indexed by C rows
            float a16[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = c_row0 + r;
                a16[r] = b2f7(hact_l[((m < kM) ? m : 0) * KS + 32]);
            }
#pragma unroll
            for (int tw = 0; tw < kTilesPerWave; ++tw) {
                const int col = (wave * kTilesPerWave + tw) * 16 + c_col;
                floatx4_t acc = {0.f, 0.f, 0.f, 0.f};
                const bf16x8_t b0 = *reinterpret_cast<const bf16x8_t*>(ud_l + col * KU + g2_k0);
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
                const float d2l = d2last_l[col];
                const float cv = c_l[col];
                const float wrv = wr_l[col];
                const float mv = mean_l[col];
                const float iv = istd_l[col];
                const bool col_in = col < O;
                float ssum = 0.0f, ssq = 0.0f;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int m = c_row0 + r;
                    const float o_new = tanh_fast(fmaf(a16[r], d2l, acc[r]) + cv);
                    fit_part[r] = fmaf(wrv, o_new, fit_part[r]);
                    if (m < live && col_in) {
                        ssum += o_new;
                        ssq = fmaf(o_new, o_new, ssq);
                    }
                    if (m < kM) {
                        const __bf16 ob = f2b7(o_new);
                        obs_l[m * OPS + col] = ob;
                        obsn_l[m * OPS + col] = f2b7((b2f7(ob) - mv) * iv);
                    }
                }
                stat_sum[tw] += ssum;
                stat_sumsq[tw] += ssq;
            }
        }
        if (!(args.skip_mask & 8)) __syncthreads();
    }

    // ---- wrap-up ----
    // Deterministic fitness reduction (no float atomics — the kernel must
    // be bitwise run-to-run reproducible): (1) shuffle-reduce fit_part
    // across the 16 lanes of each row segment, (2) stage per-wave partials
    // in LDS, (3) one thread per member sums the waves in fixed order.
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
        for (int w = 0; w < kWaves; ++w) total += wave_fit[w * 16 + tid];
        args.fitness_out[base_member + tid] =
            total + args.alive_bonus * (float)args.steps - args.act_cost * actsq_l[tid] / (float)A;
    }
    // per-block stat slice: the four 16-lane groups of a wave share each
    // col (they hold different member rows), so reduce across them with
    // fixed-order shuffles, then lanes 0-15 store — plain stores, no
    // float atomics, run-to-run deterministic
    float* stats = args.obs_stats_out + (int64_t)blockIdx.x * 2 * O;
#pragma unroll
    for (int tw = 0; tw < kTilesPerWave; ++tw) {
        float ss = stat_sum[tw], sq = stat_sumsq[tw];
        ss += __shfl_down(ss, 32, 64);
        ss += __shfl_down(ss, 16, 64);
        sq += __shfl_down(sq, 32, 64);
        sq += __shfl_down(sq, 16, 64);
        const int col = (wave * kTilesPerWave + tw) * 16 + c_col;
        if (lane < 16 && col < O) {
            stats[col] = ss;
            stats[O + col] = sq;
        }
    }
}


// ===========================================================================
// m7: MLP-64 rollout — 2-wave TEAMS per member, W1 register-resident.
//
// v6 runs MLP members with 8-lane group dots + block barriers and
// measures ~10:1 WAIT:BUSY (36 ms/gen at the reference MLP-64 config).
// Here each member is owned by a PAIR of waves: wave half h of the pair
// holds hidden rows [32h, 32h+32) of W1 in REGISTERS (32 rows x 6
// columns per lane = 96 VGPRs — a full per-member W1 cannot fit one
// wave, but half can), the observation lives in 3 bf16x2 registers per
// lane, and the layer-2 weights sit k-sliced per lane (17 bf16 = 9
// VGPRs). Per step: 32 col-sliced hidden dots per wave (DPP-reduced),
// one LDS exchange of the quantized h1 halves, the 17 action dots
// k-sliced over lanes, and the v8-style per-lane-column dynamics update
// with [U;D2] k-pairs from LDS. Two block barriers per step; all four
// members of a block run the same schedule, so barriers stay balanced.
// Numerics contract identical to v6/rollout_eager (bf16 operands, fp32
// accumulation, quantize-then-normalize, tanh on hidden and output).
// Default for the MLP-64 flagship geometry (29.4 vs v6's 36.4 ms at
// T=1000 popsize 4000, 1.24x); EVOTORCH_AMD_ROLLOUT_V6 forces v6, and
// EVOTORCH_AMD_M7_MEMBERS=4 selects the single-block 4-member variant
// (measured slower: one 8-wave block per CU exposes the barrier chain).
// ===========================================================================

template <int O, int A, int H, int kM>
__global__ __launch_bounds__(kM * 128, (kM == 2) ? 2 : 1) void rollout_m7_kernel(RolloutV7Args args) {
    constexpr int kThreads = kM * 128;           // 2 waves per member
    constexpr int OP = (O + 127) / 128 * 128;    // 384 padded cols
    constexpr int kPairs = OP / 2 / 64;          // 3 bf16x2 col-pairs per lane
    constexpr int kCols = 2 * kPairs;            // 6 cols per lane
    constexpr int HH = H / 2;                    // hidden rows per wave (32)
    constexpr int R = 16;
    constexpr int K = R + A;                     // 33 dynamics rows
    constexpr int KQ = (K + 1) / 2;              // 17 k-pairs
    constexpr int VS = OP + 8;
    constexpr int US = OP + 8;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int member_slot = wave >> 1;           // 0..3 within the block
    const int half = wave & 1;                   // which half of W1 this wave owns
    const int member = blockIdx.x * kM + member_slot;
    const bool live = member < args.n_members;
    const int mem_clamped = live ? member : 0;

    extern __shared__ unsigned char ldsm[];
    __bf16* v_l = reinterpret_cast<__bf16*>(ldsm);               // [R][VS]
    bf16x2_t* ud_l = reinterpret_cast<bf16x2_t*>(v_l + R * VS);  // [KQ][US]
    float* mean_l = reinterpret_cast<float*>(ud_l + KQ * US);    // [OP]
    float* istd_l = mean_l + OP;                                 // [OP]
    float* c_l = istd_l + OP;                                    // [OP]
    float* wr_l = c_l + OP;                                      // [OP]
    __bf16* h1_l = reinterpret_cast<__bf16*>(wr_l + OP);         // [2][kM][H] hidden (step-parity banks)
    float* act_l = reinterpret_cast<float*>(h1_l + 2 * kM * H);  // [2][kM][A + 1] actions
    float* hx_l = act_l + 2 * kM * (A + 1);                      // [2][kM][16] dynamics exchange
    float* sstat = hx_l + 2 * kM * 16;                           // [2][kM][OP] end only

    const float* e_V = args.env_blob;
    const float* e_M = args.env_blob + (size_t)R * O;
    const float* e_c = args.env_blob + (size_t)(2 * R + A) * O;
    const float* e_wr = e_c + O;
    const float* e_mean = e_wr + O;
    const float* e_std = e_mean + O;

    for (int j = tid; j < R * VS; j += kThreads) {
        const int r = j / VS, cc = j % VS;
        v_l[j] = (cc < O) ? f2b7(e_V[(size_t)r * O + cc]) : f2b7(0.0f);
    }
    for (int j = tid; j < KQ * US; j += kThreads) {
        const int q = j / US, cc = j % US;
        bf16x2_t m;
        m.x = (cc < O) ? f2b7(e_M[(size_t)(2 * q) * O + cc]) : f2b7(0.0f);
        m.y = (cc < O && 2 * q + 1 < K) ? f2b7(e_M[(size_t)(2 * q + 1) * O + cc]) : f2b7(0.0f);
        ud_l[j] = m;
    }
    for (int j = tid; j < OP; j += kThreads) {
        const bool in = j < O;
        mean_l[j] = in ? e_mean[j] : 0.0f;
        istd_l[j] = in ? 1.0f / e_std[j] : 0.0f;
        c_l[j] = in ? e_c[j] : 0.0f;
        wr_l[j] = in ? e_wr[j] : 0.0f;
    }

    // ---- per-member weights ----
    // params layout (synthetic_env.py): W1[H][O] b1[H] W2[A][H] b2[A]
    const long plen = (long)H * O + H + (long)A * H + A;
    const float* P = args.params + (long)mem_clamped * plen;
    bf16x2_t w1[HH][kPairs];  // my half's rows x my col slice
#pragma unroll
    for (int r = 0; r < HH; ++r) {
#pragma unroll
        for (int p = 0; p < kPairs; ++p) {
            // lane owns STRIDE-64 single columns (lane + 64*(2p+e)) — the
            // two columns sharing a bf16x2 are 64 apart. fdot2 only needs
            // the same pairing in both operands; adjacent-pair ownership
            // made every GEMM2 [U;D2] read stride-2 per lane (measured
            // 0.99 LDS conflict cycles per LDS instruction).
            const int cx = lane + 64 * (2 * p);
            const int cy = lane + 64 * (2 * p + 1);
            const long row = (long)(half * HH + r) * O;
            bf16x2_t w;
            w.x = (cx < O) ? f2b7(P[row + cx]) : f2b7(0.0f);
            w.y = (cy < O) ? f2b7(P[row + cy]) : f2b7(0.0f);
            w1[r][p] = w;
        }
    }
    float b1[HH];
#pragma unroll
    for (int r = 0; r < HH; ++r)
        b1[r] = __builtin_bit_cast(float, __builtin_amdgcn_readfirstlane(__builtin_bit_cast(int, P[(long)H * O + half * HH + r])));
    // W2 k-sliced: lane l holds W2[a][l] for all a (h1 index = lane)
    __bf16 w2[A];
#pragma unroll
    for (int a = 0; a < A; ++a) w2[a] = (lane < H) ? f2b7(P[(long)H * O + H + (long)a * H + lane]) : f2b7(0.0f);
    float b2[A];
#pragma unroll
    for (int a = 0; a < A; ++a)
        b2[a] = __builtin_bit_cast(float, __builtin_amdgcn_readfirstlane(__builtin_bit_cast(int, P[(long)H * O + H + (long)A * H + a])));

    float ssum[kCols] = {}, ssq[kCols] = {};
    float fit_acc = 0.0f;
    float asq_total = 0.0f;
    __syncthreads();

    // ---- initial observation ----
    const unsigned long long iseed = args.seed_ptr ? *args.seed_ptr : args.init_seed;
    bf16x2_t obs2[kPairs], obsn2[kPairs];
#pragma unroll
    for (int p = 0; p < kPairs; ++p) {
        bf16x2_t o, onr;
#pragma unroll
        for (int e = 0; e < 2; ++e) {
            const int col = lane + 64 * (2 * p + e);
            float z[4];
            philox_normal4(iseed, (uint32_t)(args.member_offset + member), (uint64_t)(col >> 2), z);
            // quad element = col & 3 = lane & 3 (64 ≡ 0 mod 4): branchless
            const float z01 = (lane & 1) ? z[1] : z[0];
            const float z23 = (lane & 1) ? z[3] : z[2];
            const float zv = (lane & 2) ? z23 : z01;
            const __bf16 ob = (col < O) ? f2b7(0.1f * zv) : f2b7(0.0f);
            const float onf = (b2f7(ob) - mean_l[col < OP ? col : 0]) * istd_l[col < OP ? col : 0];
            if (e == 0) { o.x = ob; onr.x = f2b7(onf); } else { o.y = ob; onr.y = f2b7(onf); }
        }
        obs2[p] = o;
        obsn2[p] = onr;
    }

    for (int t = 0; t < args.steps; ++t) {
        // step-parity banks let the NEXT step's phase-A writes start
        // without a loop-end barrier (2 barriers per step, not 3)
        __bf16* h1b = h1_l + (t & 1) * kM * H;
        float* actb = act_l + (t & 1) * kM * (A + 1);
        float* hxb = hx_l + (t & 1) * kM * 16;
        // ---- hidden half: 32 col-sliced dots, DPP-reduced, quantized to LDS ----
        {
            // fused dot+reduce per row: row r+1's dot2s issue underneath
            // row r's DPP chain, and no 32-deep accumulator array stays
            // live (the separated two-loop form held 32 extra VGPRs).
            // v_readlane scalarization measured as the dominant stall
            // (SALU round-trips); one cross-half shuffle puts the 64-lane
            // total in lane 31, which writes directly.
#pragma unroll
            for (int r = 0; r < HH; ++r) {
                float acc = 0.0f;
#pragma unroll
                for (int p = 0; p < kPairs; ++p) acc = __builtin_amdgcn_fdot2_f32_bf16(w1[r][p], obsn2[p], acc, false);
                float s = reduce32_dpp(acc);
                s += __shfl_xor(s, 32, 64);
                if (lane == 31)
                    h1b[member_slot * H + half * HH + r] = f2b7(tanh_fast(s + b1[r]));
            }
        }
        // ---- dynamics h = V @ obs is independent of h1: same phase, no
        // extra barrier (published into the sstat-scratch h-exchange) ----
        {
            float* hx = hxb;
#pragma unroll
            for (int rr = 0; rr < 8; ++rr) {
                const int r = half * 8 + rr;
                float acc = 0.0f;
#pragma unroll
                for (int p = 0; p < kPairs; ++p) {
                    bf16x2_t v;
                    v.x = v_l[r * VS + lane + 64 * (2 * p)];
                    v.y = v_l[r * VS + lane + 64 * (2 * p + 1)];
                    acc = __builtin_amdgcn_fdot2_f32_bf16(v, obs2[p], acc, false);
                }
                float s = reduce32_dpp(acc);
                s += __shfl_xor(s, 32, 64);
                if (lane == 31) hx[member_slot * 16 + half * 8 + rr] = s;
            }
        }
        __syncthreads();
        // ---- actions: h1 k-sliced per lane (h index = lane); the 17
        // reductions split across BOTH halves of the member's wave pair ----
        {
            const float h1v = (lane < H) ? b2f7(h1b[member_slot * H + lane]) : 0.0f;
            constexpr int kA0 = (A + 1) / 2;  // half 0: a < kA0; half 1: the rest
#pragma unroll
            for (int a = 0; a < A; ++a) {
                if ((half == 0) != (a < kA0)) continue;
                float part = b2f7(w2[a]) * h1v;
                float s = reduce32_dpp(part);
                s += __shfl_xor(s, 32, 64);
                if (lane == 31)
                    actb[member_slot * (A + 1) + a] = fminf(fmaxf(s + b2[a], -1.0f), 1.0f);
            }
        }
        __syncthreads();
        {
            bf16x2_t hact2[KQ];
            const float* hx = hxb;
#pragma unroll
            for (int q = 0; q < KQ; ++q) {
                float kv[2];
#pragma unroll
                for (int e = 0; e < 2; ++e) {
                    const int k = 2 * q + e;
                    if (k >= K) { kv[e] = 0.0f; continue; }
                    kv[e] = (k < R) ? hx[member_slot * 16 + k] : actb[member_slot * (A + 1) + (k - R)];
                }
                bf16x2_t hp;
                hp.x = f2b7(kv[0]);
                hp.y = f2b7(kv[1]);
                hact2[q] = hp;
            }
#pragma unroll
            for (int p = 0; p < kPairs; ++p) {
                bf16x2_t o, onr;
#pragma unroll
                for (int e = 0; e < 2; ++e) {
                    const int col = lane + 64 * (2 * p + e);
                    const int j = 2 * p + e;
                    float acc = 0.0f;
#pragma unroll
                    for (int q = 0; q < KQ; ++q)
                        acc = __builtin_amdgcn_fdot2_f32_bf16(ud_l[(size_t)q * US + col], hact2[q], acc, false);
                    const float o_new = tanh_fast(acc + c_l[col]);
                    if (half == 0) {  // one wave of the pair owns fitness/stats
                        fit_acc = fmaf(wr_l[col], o_new, fit_acc);
                        ssum[j] += o_new;
                        ssq[j] = fmaf(o_new, o_new, ssq[j]);
                    }
                    const __bf16 ob = f2b7(o_new);
                    const float onf = (b2f7(ob) - mean_l[col]) * istd_l[col];
                    if (e == 0) { o.x = ob; onr.x = f2b7(onf); } else { o.y = ob; onr.y = f2b7(onf); }
                }
                obs2[p] = o;
                obsn2[p] = onr;
            }
            if (half == 0 && lane < A) {
                const float av = actb[member_slot * (A + 1) + lane];
                asq_total = fmaf(av, av, asq_total);
            }
        }
        // no loop-end barrier: the next step writes the OTHER parity bank
    }
    __syncthreads();

    // ---- fitness wrap-up (wave half 0 of each member) ----
    if (half == 0) {
        float f = reduce32_dpp(fit_acc);
        float total = __builtin_bit_cast(float, __builtin_amdgcn_readlane(__builtin_bit_cast(int, f), 31)) +
                      __builtin_bit_cast(float, __builtin_amdgcn_readlane(__builtin_bit_cast(int, f), 63));
        float aq = reduce32_dpp(asq_total);
        float aq_total = __builtin_bit_cast(float, __builtin_amdgcn_readlane(__builtin_bit_cast(int, aq), 31)) +
                         __builtin_bit_cast(float, __builtin_amdgcn_readlane(__builtin_bit_cast(int, aq), 63));
        if (lane == 0 && live)
            args.fitness_out[member] =
                total + args.alive_bonus * (float)args.steps - args.act_cost * aq_total / (float)A;
    }
    __syncthreads();
    // ---- block stat partial (sstat was scratch during the loop; rebuilt now) ----
    if (half == 0) {  // one writer per member slot (the stats-owning wave)
#pragma unroll
        for (int p = 0; p < kPairs; ++p) {
#pragma unroll
            for (int e = 0; e < 2; ++e) {
                const int col = lane + 64 * (2 * p + e);
                if (col < OP) {
                    sstat[(size_t)member_slot * OP + col] = live ? ssum[2 * p + e] : 0.0f;
                    sstat[(size_t)(kM + member_slot) * OP + col] = live ? ssq[2 * p + e] : 0.0f;
                }
            }
        }
    }
    __syncthreads();
    float* stats = args.obs_stats_out + (int64_t)blockIdx.x * 2 * O;
    for (int col = tid; col < O; col += kThreads) {
        float s = 0.0f, q = 0.0f;
#pragma unroll
        for (int w = 0; w < kM; ++w) {
            s += sstat[(size_t)w * OP + col];
            q += sstat[(size_t)(kM + w) * OP + col];
        }
        stats[col] = s;
        stats[O + col] = q;
    }
}

template <int O_T, int A_T, int H_T, int kM_T>
static void launch_m7(const RolloutV7Args& args, int n, hipStream_t stream) {
    constexpr int OP = (O_T + 127) / 128 * 128;
    constexpr int KQ = (16 + A_T + 1) / 2;
    const size_t lds = (size_t)16 * (OP + 8) * 2 + (size_t)KQ * (OP + 8) * 4 + (size_t)4 * OP * 4 +
                       (size_t)2 * kM_T * H_T * 2 + (size_t)2 * kM_T * (A_T + 1) * 4 +
                       (size_t)2 * kM_T * 16 * 4 + (size_t)2 * kM_T * OP * 4;
    static bool attr_m7[2] = {false, false};
    const int slot = (kM_T == 2) ? 0 : 1;
    if (!attr_m7[slot]) {
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&rollout_m7_kernel<O_T, A_T, H_T, kM_T>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_m7[slot] = true;
    }
    const int blocks = (n + kM_T - 1) / kM_T;
    hipLaunchKernelGGL((rollout_m7_kernel<O_T, A_T, H_T, kM_T>), dim3(blocks), dim3(kM_T * 128), lds, stream, args);
}

template <int O_T, int A_T, int kWaves>
static void launch_v7(const RolloutV7Args& args, int n, hipStream_t stream) {
    constexpr int OP = (O_T + 127) / 128 * 128;
    constexpr int kM = 2 * kWaves;
    size_t lds = (size_t)(2 * kM * (OP + 8) + 16 * 72 + 16 * (OP + 8) + OP * 40) * 2 +
                 (size_t)(kM * A_T + 5 * OP + kWaves * 16 + kM) * 4;
    static bool attr_set7[2] = {false, false};
    const int slot = (kWaves == 8) ? 1 : 0;
    if (!attr_set7[slot]) {
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&rollout_v7_kernel<O_T, A_T, kWaves>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set7[slot] = true;
    }
    const int blocks = (n + kM - 1) / kM;
    hipLaunchKernelGGL((rollout_v7_kernel<O_T, A_T, kWaves>), dim3(blocks), dim3(64 * kWaves), lds, stream, args);
}

void rollout_v7(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out, torch::Tensor fitness,
                int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus, double act_cost,
                int64_t init_seed, int64_t member_offset, const unsigned long long* seed_ptr) {
    const int n = (int)params.size(0);
    const int O = (int)obs_dim, A = (int)act_dim, R = (int)rank;
    TORCH_CHECK(R == 16, "rollout v7 requires rank 16");

    RolloutV7Args args;
    args.seed_ptr = seed_ptr;
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

    constexpr int O_T = 376, A_T = 17;
    TORCH_CHECK(O == O_T && A == A_T, "rollout v7 instantiated for the Humanoid geometry (obs 376, act 17)");
    auto stream = at::cuda::getCurrentCUDAStream();
    args.skip_mask = 0;
    if (const char* skp = getenv("EVOTORCH_AMD_V7_SKIP")) args.skip_mask = atoi(skp);  // perf probe only
    const char* wenv = getenv("EVOTORCH_AMD_ROLLOUT_V7_WAVES");
    if (wenv && atoi(wenv) == 4) {
        launch_v7<O_T, A_T, 4>(args, n, stream);
    } else {
        launch_v7<O_T, A_T, 8>(args, n, stream);
    }
}


void rollout_m7(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out, torch::Tensor fitness,
                int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus, double act_cost,
                int64_t init_seed, int64_t member_offset, const unsigned long long* seed_ptr) {
    const int n = (int)params.size(0);
    TORCH_CHECK((int)rank == 16 && (int)obs_dim == 376 && (int)act_dim == 17,
                "rollout m7 is instantiated for the Humanoid geometry");
    RolloutV7Args args;
    args.params = params.data_ptr<float>();
    args.env_blob = env_blob.data_ptr<float>();
    args.fitness_out = fitness.data_ptr<float>();
    args.obs_stats_out = obs_stats_out.data_ptr<float>();
    args.n_members = n;
    args.member_offset = (long)member_offset;
    args.obs_dim = (int)obs_dim; args.act_dim = (int)act_dim; args.rank = (int)rank;
    args.steps = (int)steps;
    args.alive_bonus = (float)alive_bonus;
    args.act_cost = (float)act_cost;
    args.init_seed = (unsigned long long)init_seed;
    args.seed_ptr = seed_ptr;
    args.skip_mask = 0;
    auto stream = at::cuda::getCurrentCUDAStream();
    const char* km = getenv("EVOTORCH_AMD_M7_MEMBERS");
    if (km && atoi(km) == 4) {
        launch_m7<376, 17, 64, 4>(args, n, stream);
    } else {
        launch_m7<376, 17, 64, 2>(args, n, stream);  // 2 blocks/CU cover stalls
    }
}

}  // namespace ea
