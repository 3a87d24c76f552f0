// K10+K11 fused: whole-generation rollout of a population of policies
// through the synthetic vectorized environment, one kernel launch per
// generation (SURVEY.md §3.4 — the VecGymNE hot loop, collapsed).
//
// Policies: linear (obs -> act) or one-hidden-layer tanh MLP
// (obs -> H -> act; H=64 is the reference paper's brax-humanoid config).
//
// MI355X design (v6): one workgroup rolls out kMembers (default 2)
// population members for the entire T-step episode with policy weights
// AND the shared environment matrices resident in LDS — the inner loop
// never touches global memory. Multi-member blocks exist because one
// member's per-step work underfills a 512-thread block; gfx950 allows the
// required LDS (sharedMemPerBlock = 160 KB, probed; ~75 KB for 2 linear
// members, ~146 KB for 2 MLP-64 members).
//
// All inner products run on `v_dot2_f32_bf16` (2 bf16 MACs/instruction,
// fp32 accumulate). LDS access patterns:
//   * group-reduced dots (policy layers, dynamics factor V): row-major
//     [out][len]; 8-lane groups read consecutive bf16x2 — conflict-free,
//     with a 3-level shuffle reduction (64-lane trees serialized 6
//     dependent shuffles; measured 10:1 SQ_WAIT:SQ_BUSY).
//   * per-thread dots (U, D2): PAIR-INTERLEAVED COLUMN-MAJOR [K/2][O]
//     bf16x2 — one 4 B read per dot2 at lane-consecutive addresses.
// Observation-normalization statistics (sum, sumsq) accumulate per-thread
// in registers and merge with one atomic pass at the end (K11; they
// become a single RCCL all-reduce across ranks — SURVEY.md §2.8 P5).
//
// Environment spec (must match the eager reference in
// evotorch_amd/neuroevolution/synthetic_env.py::rollout_eager, which
// quantizes the same operands to bf16):
//   obs_n = bf16((obs − mean) · inv_std)
//   hid   = bf16(tanh(W1·obs_n + b1))            (MLP mode only)
//   a     = clip(W2·hid + b2, −1, 1)   |  clip(W·obs_n + b, −1, 1)
//   h     = bf16(V·obs)
//   o'    = tanh(Σ_i U_T[i]·h[i] + Σ_m D2_T[m]·a[m] + c)
//   r     = wr·o' + alive_bonus − act_cost·‖a‖²/A
//   o₀    = 0.1·philox_normal(member);  fitness = Σ_t r_t

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdlib>

#include "philox.h"
#include "reduce.h"

namespace ea {

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a ROCm tensor")

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

__device__ __forceinline__ __bf16 f2b(float v) { return (__bf16)v; }
__device__ __forceinline__ float b2f(__bf16 v) { return (float)v; }

// tanh via the hardware exp unit (~1e-6 rel error, far below the bf16
// quantization of every operand; libm tanhf is branchy on the hot path)
// 8-lane group sum on the VALU pipe: after row_shr 1/2/4 the TOP lane of
// each 8-lane group (glane == 7) holds the group sum (__shfl_down lowers
// to ds_bpermute_b32 on the LDS pipe — the v7 kernel measured that
// contention; same fix here).
template <int kCtrl6>
__device__ __forceinline__ float dpp_add6(float x) {
    const int moved = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, x), kCtrl6, 0xf, 0xf, true);
    return x + __builtin_bit_cast(float, moved);
}

__device__ __forceinline__ float group8_sum_dpp(float x) {
    x = dpp_add6<0x111>(x);  // row_shr:1
    x = dpp_add6<0x112>(x);  // row_shr:2
    x = dpp_add6<0x114>(x);  // row_shr:4
    return x;                // valid in the top lane of each 8-lane group
}

__device__ __forceinline__ float tanh_fast6(float x) {
    const float xc = fminf(fmaxf(x, -15.0f), 15.0f);
    const float e = __expf(2.0f * xc);
    return (e - 1.0f) * __builtin_amdgcn_rcpf(e + 1.0f);
}

// runtime member-index -> pointer select WITHOUT a runtime-indexed array
// (which would demote to scratch memory — common-mistake #20); for
// kMembers<=2 this folds to a single v_cndmask per operand.
template <typename T>
__device__ __forceinline__ T* sel2(int m, T* p0, T* p1) {
    return m == 0 ? p0 : p1;
}

struct RolloutArgs {
    const float* params;      // [n_members][L]
    const float* env_blob;    // packed fp32 env data (see env_blob())
    float* fitness_out;       // [n_members]
    float* obs_stats_out;     // [2][O]  (sum, sumsq) — atomically accumulated
    int n_members;
    long member_offset;       // global id of member 0 (rank sharding)
    int obs_dim, act_dim, rank, steps, hidden;
    float alive_bonus, act_cost;
    unsigned long long init_seed;
    const unsigned long long* seed_ptr;  // device episode seed (hipGraph-safe); overrides init_seed
};

// one group-reduced dot output: 8 lanes compute row·vec, lane 0 of the
// group applies the epilogue selected by `kind`
struct OutDesc {
    const __bf16* row;
    const __bf16* vec;
    __bf16* dst;
    const float* bias;   // null for kind 0
    int kind;            // 0 = plain store, 1 = action (bias+clip), 2 = hidden (bias+tanh)
    int member;
    bool valid;
};

// env_blob layout (fp32): V [R][O] · U_T [R][O] · D2_T [A][O] · c [O] ·
// wr [O] · mean [O] · std [O]

template <int kGroup, int kMembers>
__global__ __launch_bounds__(512, 2) void rollout_linear_kernel(RolloutArgs args) {
    const int O = args.obs_dim, A = args.act_dim, R = args.rank, H = args.hidden;
    const int tid = threadIdx.x;
    const int base_member = blockIdx.x * kMembers;
    if (base_member >= args.n_members) return;
    const int live_members = min(kMembers, args.n_members - base_member);
    const int A_PAD = (A + 2) & ~1;
    const int R_PAD = (R + 2) & ~1;
    const long param_len = (H > 0) ? ((long)H * O + H + (long)A * H + A) : ((long)A * O + A);

    extern __shared__ unsigned char lds_raw[];
    // ---- shared environment ----
    __bf16* V_l = reinterpret_cast<__bf16*>(lds_raw);          // [R][O]
    bf16x2* U_pair = reinterpret_cast<bf16x2*>(V_l + R * O);   // [R_PAD/2][O]
    bf16x2* D2_pair = U_pair + (R_PAD / 2) * O;                // [A_PAD/2][O]
    float* c_l = reinterpret_cast<float*>(D2_pair + (A_PAD / 2) * O);  // [O]
    float* wr_l = c_l + O;                                     // [O]
    float* mean_l = wr_l + O;                                  // [O]
    float* istd_l = mean_l + O;                                // [O]
    float* scratch = istd_l + O;                               // [16]
    // ---- per-member state ----
    __bf16* W1_l[kMembers];   // linear: [A][O]; MLP: [H][O]
    float* b1_l[kMembers];    // linear: [A];    MLP: [H]
    __bf16* W2_l[kMembers];   // MLP only: [A][H]
    float* b2_l[kMembers];    // MLP only: [A]
    __bf16* hid_b[kMembers];  // MLP only: [H]
    __bf16* obs_b[kMembers];
    __bf16* obsn_b[kMembers];
    __bf16* h_b[kMembers];
    __bf16* act_b[kMembers];
    {
        unsigned char* cursor = reinterpret_cast<unsigned char*>(scratch + 16);
#pragma unroll
        for (int m = 0; m < kMembers; ++m) {
            const int w1_rows = (H > 0) ? H : A;
            W1_l[m] = reinterpret_cast<__bf16*>(cursor);
            cursor += (size_t)w1_rows * O * 2;
            b1_l[m] = reinterpret_cast<float*>(cursor);
            cursor += (size_t)w1_rows * 4;
            if (H > 0) {
                W2_l[m] = reinterpret_cast<__bf16*>(cursor);
                cursor += (size_t)A * H * 2;
                b2_l[m] = reinterpret_cast<float*>(cursor);
                cursor += (size_t)A * 4;
                hid_b[m] = reinterpret_cast<__bf16*>(cursor);
                cursor += (size_t)H * 2;
            } else {
                W2_l[m] = nullptr;
                b2_l[m] = nullptr;
                hid_b[m] = nullptr;
            }
            obs_b[m] = reinterpret_cast<__bf16*>(cursor);
            cursor += (size_t)O * 2;
            obsn_b[m] = reinterpret_cast<__bf16*>(cursor);
            cursor += (size_t)O * 2;
            h_b[m] = reinterpret_cast<__bf16*>(cursor);
            cursor += (size_t)R_PAD * 2;
            act_b[m] = reinterpret_cast<__bf16*>(cursor);
            cursor += (size_t)A_PAD * 2;
        }
    }

    // ---- stage shared env ----
    {
        const float* e = args.env_blob;
        const int RO = R * O, AO = A * O;
        for (int i = tid; i < RO; i += blockDim.x) V_l[i] = f2b(e[i]);
        for (int i = tid; i < (R_PAD / 2) * O; i += blockDim.x) {
            const int i2 = i / O, j = i % O;
            bf16x2 v2;
            v2.x = (2 * i2 < R) ? f2b(e[RO + (2 * i2) * O + j]) : f2b(0.0f);
            v2.y = (2 * i2 + 1 < R) ? f2b(e[RO + (2 * i2 + 1) * O + j]) : f2b(0.0f);
            U_pair[i2 * O + j] = v2;
        }
        for (int i = tid; i < (A_PAD / 2) * O; i += blockDim.x) {
            const int m2 = i / O, j = i % O;
            bf16x2 v2;
            v2.x = (2 * m2 < A) ? f2b(e[2 * RO + (2 * m2) * O + j]) : f2b(0.0f);
            v2.y = (2 * m2 + 1 < A) ? f2b(e[2 * RO + (2 * m2 + 1) * O + j]) : f2b(0.0f);
            D2_pair[m2 * O + j] = v2;
        }
        const float* tail = e + 2 * RO + AO;
        for (int j = tid; j < O; j += blockDim.x) {
            c_l[j] = tail[j];
            wr_l[j] = tail[O + j];
            mean_l[j] = tail[2 * O + j];
            istd_l[j] = 1.0f / tail[3 * O + j];
        }
    }
    // ---- stage members: weights, pads, initial observations ----
#pragma unroll
    for (int m = 0; m < kMembers; ++m) {
        if (m >= live_members) break;
        const float* my_params = args.params + (long)(base_member + m) * param_len;
        if (H > 0) {
            for (int i = tid; i < H * O; i += blockDim.x) W1_l[m][i] = f2b(my_params[i]);
            for (int i = tid; i < H; i += blockDim.x) b1_l[m][i] = my_params[H * O + i];
            for (int i = tid; i < A * H; i += blockDim.x) W2_l[m][i] = f2b(my_params[H * O + H + i]);
            for (int i = tid; i < A; i += blockDim.x) b2_l[m][i] = my_params[H * O + H + A * H + i];
        } else {
            for (int i = tid; i < A * O; i += blockDim.x) W1_l[m][i] = f2b(my_params[i]);
            for (int i = tid; i < A; i += blockDim.x) b1_l[m][i] = my_params[A * O + i];
        }
        for (int i = tid; i < R_PAD; i += blockDim.x) h_b[m][i] = f2b(0.0f);
        for (int i = tid; i < A_PAD; i += blockDim.x) act_b[m][i] = f2b(0.0f);
        const unsigned long long gmember = (unsigned long long)(args.member_offset + base_member + m);
        const unsigned long long iseed = args.seed_ptr ? *args.seed_ptr : args.init_seed;
        for (int j4 = tid; j4 * 4 < O; j4 += blockDim.x) {
            float z[4];
            philox_normal4(iseed, (uint32_t)gmember, (uint64_t)j4, z);
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                const int j = j4 * 4 + u;
                if (j < O) obs_b[m][j] = f2b(0.1f * z[u]);
            }
        }
    }
    __syncthreads();

    const int lane = tid & (kWaveSize - 1);
    const int wave = tid / kWaveSize;
    const int group = lane / kGroup;
    const int glane = lane % kGroup;
    const int groups_per_block = (int)(blockDim.x / kGroup);
    const int groups_per_wave = kWaveSize / kGroup;
    const int my_group = wave * groups_per_wave + group;

    // ---- phase-A table: dots over obs-side vectors -------------------------
    // Per member: H hidden outputs (MLP) or A action outputs (linear),
    // followed by R dynamics-factor outputs. Up to 3 rounds (e.g. 2 MLP-64
    // members: 2*(64+16) = 160 outputs over 64 groups).
    constexpr int kRoundsA = 3;
    const int a_per_member = (H > 0 ? H : A) + R;
    const int n_outs_a = live_members * a_per_member;
    OutDesc descA[kRoundsA];
#pragma unroll
    for (int r = 0; r < kRoundsA; ++r) {
        const int out = my_group + r * groups_per_block;
        OutDesc d;
        d.valid = out < n_outs_a;
        const int safe_out = d.valid ? out : 0;
        const int m = safe_out / a_per_member;
        const int lo = safe_out % a_per_member;
        d.member = m;
        const int first_count = (H > 0 ? H : A);
        __bf16* w1_m = sel2(m, W1_l[0], W1_l[kMembers - 1]);
        float* b1_m = sel2(m, b1_l[0], b1_l[kMembers - 1]);
        __bf16* obs_m = sel2(m, obs_b[0], obs_b[kMembers - 1]);
        __bf16* obsn_m = sel2(m, obsn_b[0], obsn_b[kMembers - 1]);
        __bf16* henv_m = sel2(m, h_b[0], h_b[kMembers - 1]);
        if (lo < first_count) {
            d.row = w1_m + lo * O;
            d.vec = obsn_m;
            d.bias = b1_m + lo;
            if (H > 0) {
                __bf16* hid_m = sel2(m, hid_b[0], hid_b[kMembers - 1]);
                d.dst = hid_m + lo;
                d.kind = 2;
            } else {
                __bf16* act_m = sel2(m, act_b[0], act_b[kMembers - 1]);
                d.dst = act_m + lo;
                d.kind = 1;
            }
        } else {
            d.row = V_l + (lo - first_count) * O;
            d.vec = obs_m;
            d.bias = nullptr;
            d.dst = henv_m + (lo - first_count);
            d.kind = 0;
        }
        descA[r] = d;
    }
    // ---- phase-B table (MLP only): action dots over the hidden vector ------
    constexpr int kRoundsB = 1;
    OutDesc descB[kRoundsB];
#pragma unroll
    for (int r = 0; r < kRoundsB; ++r) {
        const int out = my_group + r * groups_per_block;
        OutDesc d;
        d.valid = (H > 0) && (out < live_members * A);
        const int safe_out = d.valid ? out : 0;
        const int m = safe_out / A;
        const int lo = safe_out % A;
        d.member = m;
        d.row = (H > 0) ? sel2(m, W2_l[0], W2_l[kMembers - 1]) + lo * H : nullptr;
        d.vec = (H > 0) ? sel2(m, hid_b[0], hid_b[kMembers - 1]) : nullptr;
        d.bias = (H > 0) ? sel2(m, b2_l[0], b2_l[kMembers - 1]) + lo : nullptr;
        d.dst = sel2(m, act_b[0], act_b[kMembers - 1]) + lo;
        d.kind = 1;
        descB[r] = d;
    }

    float fit_part[kMembers];
    float actsq_part[kMembers];
#pragma unroll
    for (int m = 0; m < kMembers; ++m) {
        fit_part[m] = 0.0f;
        actsq_part[m] = 0.0f;
    }
    float stat_sum[2] = {0.0f, 0.0f}, stat_sumsq[2] = {0.0f, 0.0f};

    // initial normalization (later steps fuse it into phase 2's epilogue)
    for (int j = tid; j < live_members * O; j += blockDim.x) {
        const int m = j / O, jo = j % O;
        __bf16* obs_m = sel2(m, obs_b[0], obs_b[kMembers - 1]);
        __bf16* obsn_m = sel2(m, obsn_b[0], obsn_b[kMembers - 1]);
        obsn_m[jo] = f2b((b2f(obs_m[jo]) - mean_l[jo]) * istd_l[jo]);
    }
    __syncthreads();

    // group-dot worker: acc[r] = descs[r].row · descs[r].vec  (length
    // 2*n_pairs), reduced across the kGroup lanes of the group
#define GROUP_DOTS(descs, NR, n_pairs_expr, acc)                                              \
    {                                                                                         \
        const int np = (n_pairs_expr);                                                        \
        _Pragma("unroll") for (int r = 0; r < (NR); ++r) acc[r] = 0.0f;                       \
        _Pragma("unroll") for (int r = 0; r < (NR); ++r) {                                    \
            if (!descs[r].valid) continue;                                                    \
            const __bf16* row = descs[r].row;                                                 \
            const __bf16* vec = descs[r].vec;                                                 \
            float a0 = 0.0f, a1 = 0.0f;                                                       \
            const int full_iters = np / kGroup;                                               \
            const int tail = np % kGroup;                                                     \
            for (int i = 0; i + 1 < full_iters; i += 2) {                                     \
                const int p0 = glane + i * kGroup;                                            \
                const int p1 = glane + (i + 1) * kGroup;                                      \
                a0 = __builtin_amdgcn_fdot2_f32_bf16(                                         \
                    *reinterpret_cast<const bf16x2*>(row + 2 * p0),                           \
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p0), a0, false);               \
                a1 = __builtin_amdgcn_fdot2_f32_bf16(                                         \
                    *reinterpret_cast<const bf16x2*>(row + 2 * p1),                           \
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p1), a1, false);               \
            }                                                                                 \
            if (full_iters & 1) {                                                             \
                const int p0 = glane + (full_iters - 1) * kGroup;                             \
                a0 = __builtin_amdgcn_fdot2_f32_bf16(                                         \
                    *reinterpret_cast<const bf16x2*>(row + 2 * p0),                           \
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p0), a0, false);               \
            }                                                                                 \
            if (tail && glane < tail) {                                                       \
                const int p0 = glane + full_iters * kGroup;                                   \
                a1 = __builtin_amdgcn_fdot2_f32_bf16(                                         \
                    *reinterpret_cast<const bf16x2*>(row + 2 * p0),                           \
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p0), a1, false);               \
            }                                                                                 \
            acc[r] = a0 + a1;                                                                 \
        }                                                                                     \
        _Pragma("unroll") for (int r = 0; r < (NR); ++r) {                                    \
            /* VALU DPP group reduce; group sums land in glane == kGroup-1 */                 \
            if (descs[r].valid) acc[r] = group8_sum_dpp(acc[r]);                              \
        }                                                                                     \
    }

#define DOT_EPILOGUE(descs, NR, acc)                                                          \
    if (glane == kGroup - 1) {                                                                \
        _Pragma("unroll") for (int r = 0; r < (NR); ++r) {                                    \
            if (!descs[r].valid) continue;                                                    \
            if (descs[r].kind == 1) {                                                         \
                const float a = fminf(fmaxf(acc[r] + *descs[r].bias, -1.0f), 1.0f);           \
                *descs[r].dst = f2b(a);                                                       \
                _Pragma("unroll") for (int m = 0; m < kMembers; ++m) {                        \
                    if (descs[r].member == m) actsq_part[m] = fmaf(a, a, actsq_part[m]);      \
                }                                                                             \
            } else if (descs[r].kind == 2) {                                                  \
                *descs[r].dst = f2b(tanh_fast6(acc[r] + *descs[r].bias));                          \
            } else {                                                                          \
                *descs[r].dst = f2b(acc[r]);                                                  \
            }                                                                                 \
        }                                                                                     \
    }

    for (int t = 0; t < args.steps; ++t) {
        // ---- phase A: hidden (or action) + dynamics-factor dots ----
        float accA[kRoundsA];
        GROUP_DOTS(descA, kRoundsA, O / 2, accA);
        DOT_EPILOGUE(descA, kRoundsA, accA);
        __syncthreads();

        // ---- phase B (MLP only): action dots over the hidden vector ----
        if (H > 0) {
            float accB[kRoundsB];
            GROUP_DOTS(descB, kRoundsB, H / 2, accB);
            DOT_EPILOGUE(descB, kRoundsB, accB);
            __syncthreads();
        }

        // ---- dynamics phase: per-thread dims, norm fused in epilogue ----
        for (int j = tid; j < live_members * O; j += blockDim.x) {
            const int m = j / O, jo = j % O;
            float uacc = c_l[jo], dacc = 0.0f;
            const __bf16* hv = sel2(m, h_b[0], h_b[kMembers - 1]);
            const __bf16* av = sel2(m, act_b[0], act_b[kMembers - 1]);
#pragma unroll
            for (int p = 0; p < R_PAD / 2; ++p) {
                uacc = __builtin_amdgcn_fdot2_f32_bf16(
                    U_pair[p * O + jo], *reinterpret_cast<const bf16x2*>(hv + 2 * p), uacc, false);
            }
#pragma unroll
            for (int p = 0; p < A_PAD / 2; ++p) {
                dacc = __builtin_amdgcn_fdot2_f32_bf16(
                    D2_pair[p * O + jo], *reinterpret_cast<const bf16x2*>(av + 2 * p), dacc, false);
            }
            const float o_new = tanh_fast6(uacc + dacc);
#pragma unroll
            for (int mm = 0; mm < kMembers; ++mm) {
                if (m == mm) fit_part[mm] = fmaf(wr_l[jo], o_new, fit_part[mm]);
            }
            const int slot = j >= (int)blockDim.x;
            stat_sum[slot] += o_new;
            stat_sumsq[slot] = fmaf(o_new, o_new, stat_sumsq[slot]);
            const __bf16 ob = f2b(o_new);
            __bf16* obs_m = sel2(m, obs_b[0], obs_b[kMembers - 1]);
            __bf16* obsn_m = sel2(m, obsn_b[0], obsn_b[kMembers - 1]);
            obs_m[jo] = ob;
            obsn_m[jo] = f2b((b2f(ob) - mean_l[jo]) * istd_l[jo]);
        }
        __syncthreads();
    }

#undef GROUP_DOTS
#undef DOT_EPILOGUE

    // ---- wrap-up: per-member fitness reductions + obs-stat atomics ----
#pragma unroll
    for (int m = 0; m < kMembers; ++m) {
        if (m >= live_members) break;
        const float total = block_reduce_sum<false>(fit_part[m], scratch);
        __syncthreads();
        const float act_total = block_reduce_sum<false>(actsq_part[m], scratch);
        __syncthreads();
        if (tid == 0) {
            args.fitness_out[base_member + m] =
                total + args.alive_bonus * args.steps - args.act_cost * act_total / (float)A;
        }
    }
    // per-(block, member) stat slices: the j-loop visits each (m, jo) pair
    // exactly once, so plain stores are race-free and — unlike the float
    // atomicAdd they replace — run-to-run deterministic; the launcher sums
    // the partial slices with one torch reduction.
    for (int j = tid; j < live_members * O; j += blockDim.x) {
        const int slot = j >= (int)blockDim.x;
        const int m = j / O;
        const int jo = j % O;
        float* stats = args.obs_stats_out + ((int64_t)blockIdx.x * kMembers + m) * 2 * O;
        stats[jo] = stat_sum[slot];
        stats[O + jo] = stat_sumsq[slot];
    }
}

template <int kGroup, int kMembers>
static void launch_rollout(int n_blocks, int block, size_t lds_bytes, hipStream_t stream, const RolloutArgs& args) {
    static bool attr_set = false;
    if (!attr_set && lds_bytes > 64 * 1024) {
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&rollout_linear_kernel<kGroup, kMembers>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set = true;
    }
    hipLaunchKernelGGL((rollout_linear_kernel<kGroup, kMembers>), dim3(n_blocks), dim3(block), lds_bytes, stream, args);
}

void rollout_v7(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out, torch::Tensor fitness,
                int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus, double act_cost,
                int64_t init_seed, int64_t member_offset, const unsigned long long* seed_ptr);  // rollout_v7.hip
void rollout_m7(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out, torch::Tensor fitness,
                int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus, double act_cost,
                int64_t init_seed, int64_t member_offset, const unsigned long long* seed_ptr);  // rollout_v7.hip

torch::Tensor rollout_linear(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out,
                             int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus,
                             double act_cost, int64_t init_seed, int64_t member_offset, int64_t policy_hidden,
                             c10::optional<torch::Tensor> seed_buf) {
    const unsigned long long* seed_ptr = nullptr;
    if (seed_buf.has_value()) {
        TORCH_CHECK(seed_buf->is_cuda() && seed_buf->scalar_type() == at::ScalarType::Long && seed_buf->numel() >= 1,
                    "seed_buf must be a device int64 scalar");
        seed_ptr = reinterpret_cast<const unsigned long long*>(seed_buf->data_ptr<int64_t>());
    }
    CHECK_GPU(params);
    TORCH_CHECK(params.is_contiguous() && params.dim() == 2, "params must be contiguous [N][L]");
    TORCH_CHECK(params.scalar_type() == at::ScalarType::Float, "params must be fp32");
    const int n = (int)params.size(0);
    const int O = (int)obs_dim, A = (int)act_dim, R = (int)rank, H = (int)policy_hidden;
    const int64_t expected_len = (H > 0) ? ((int64_t)H * O + H + (int64_t)A * H + A) : ((int64_t)A * O + A);
    TORCH_CHECK(params.size(1) == expected_len, "param length mismatch: got ", params.size(1), " expected ", expected_len);
    TORCH_CHECK(O % 2 == 0, "obs_dim must be even (bf16x2 packing)");
    TORCH_CHECK(H % 2 == 0, "policy_hidden must be even (bf16x2 packing)");
    auto fitness = torch::empty({n}, params.options());
    // v7 (MFMA, 16 members/block) serves the linear flagship geometry;
    // v6 covers MLP policies and off-geometry envs.
    if (H == 64 && R == 16 && O == 376 && A == 17 && !getenv("EVOTORCH_AMD_ROLLOUT_V6")) {
        const char* km_env = getenv("EVOTORCH_AMD_M7_MEMBERS");
        const int m7_members = (km_env && atoi(km_env) == 4) ? 4 : 2;
        const int n_pblocks = (n + m7_members - 1) / m7_members;
        auto stat_partials = torch::zeros({(int64_t)n_pblocks, 2 * (int64_t)O}, params.options());
        rollout_m7(params, env_blob, stat_partials, fitness, obs_dim, act_dim, rank, steps, alive_bonus, act_cost,
                   init_seed, member_offset, seed_ptr);
        obs_stats_out.add_(stat_partials.sum(0));
        return fitness;
    }
    if (H == 0 && R == 16 && O == 376 && A == 17 && !getenv("EVOTORCH_AMD_ROLLOUT_V6")) {
        const int n_blocks7 = (n + 15) / 16;
        auto stat_partials = torch::zeros({(int64_t)n_blocks7, 2 * (int64_t)O}, params.options());
        rollout_v7(params, env_blob, stat_partials, fitness, obs_dim, act_dim, rank, steps, alive_bonus, act_cost,
                   init_seed, member_offset, seed_ptr);
        obs_stats_out.add_(stat_partials.sum(0));
        return fitness;
    }

    RolloutArgs args;
    args.params = params.data_ptr<float>();
    args.env_blob = env_blob.data_ptr<float>();
    args.fitness_out = fitness.data_ptr<float>();
    args.obs_stats_out = obs_stats_out.data_ptr<float>();
    args.n_members = n;
    args.member_offset = (long)member_offset;
    args.obs_dim = O; args.act_dim = A; args.rank = R; args.hidden = H;
    args.steps = (int)steps;
    args.alive_bonus = (float)alive_bonus;
    args.act_cost = (float)act_cost;
    args.init_seed = (unsigned long long)init_seed;
    args.seed_ptr = seed_ptr;

    const int block = 512;
    const int A_PAD = (A + 2) & ~1, R_PAD = (R + 2) & ~1;
    const size_t shared_bytes = ((size_t)R * O + (size_t)R_PAD * O + (size_t)A_PAD * O) * 2 + 4 * (size_t)O * 4 + 16 * 4 + 64;
    const int w1_rows = (H > 0) ? H : A;
    size_t member_bytes = ((size_t)w1_rows * O + 2 * (size_t)O + R_PAD + A_PAD) * 2 + (size_t)w1_rows * 4;
    if (H > 0) member_bytes += (size_t)A * H * 2 + (size_t)A * 4 + (size_t)H * 2;

    int members = 2;
    if (const char* env = getenv("EVOTORCH_AMD_ROLLOUT_MEMBERS")) members = atoi(env);
    TORCH_CHECK(members == 1 || members == 2, "EVOTORCH_AMD_ROLLOUT_MEMBERS must be 1 or 2");
    if (shared_bytes + 2 * member_bytes > 160 * 1024) members = 1;
    const size_t lds_bytes = shared_bytes + (size_t)members * member_bytes;
    TORCH_CHECK(lds_bytes <= 160 * 1024, "rollout LDS footprint too large: ", lds_bytes, " bytes");
    // phase-A round capacity: up to 3 rounds of (threads/8) outputs
    TORCH_CHECK(members * ((H > 0 ? H : A) + R) <= 3 * block / 8, "policy too wide for the 3-round phase-A table");
    TORCH_CHECK(members * A <= block / 8, "act_dim too large for the 1-round phase-B table");
    TORCH_CHECK(O * members <= 2 * block, "obs_dim too large for the 2-slot stat accumulators");

    auto stream = at::cuda::getCurrentCUDAStream();
    const int n_blocks = (n + members - 1) / members;
    auto stat_partials = torch::zeros({(int64_t)n_blocks * members, 2 * (int64_t)O}, params.options());
    args.obs_stats_out = stat_partials.data_ptr<float>();
    if (members == 2) {
        launch_rollout<8, 2>(n_blocks, block, lds_bytes, stream, args);
    } else {
        launch_rollout<8, 1>(n_blocks, block, lds_bytes, stream, args);
    }
    obs_stats_out.add_(stat_partials.sum(0));
    return fitness;
}

}  // namespace ea
