// K10+K11 fused: whole-generation rollout of a population of linear
// policies through the synthetic vectorized environment, one kernel launch
// per generation (SURVEY.md §3.4 — the VecGymNE hot loop, collapsed).
//
// MI355X design (v2): one workgroup per population member. The member's
// policy weights AND the (shared) environment matrices are staged once
// into LDS and stay resident for the entire T-step episode — the inner
// loop touches no global memory at all. All matrix operands are bf16 and
// every inner product runs on `v_dot2_f32_bf16` (2 bf16 MACs/instruction,
// fp32 accumulate, no per-element converts). LDS layouts:
//   * wave-reduced dots (policy W, dynamics factor V): row-major [out][O],
//     lanes read consecutive bf16x2 along O — conflict-free.
//   * per-thread dots (U, D2): per-OUTPUT rows padded to stride 18
//     (36 B ⇒ bank index advances by 9 per lane, gcd(9,32)=1 ⇒ ≤2-way,
//     i.e. free on CDNA4 — see cdna_hip_programming.md §6 G4).
// Observation-normalization statistics (sum, sumsq) accumulate per-thread
// in registers and are merged with one atomic pass at the end (K11; they
// become a single RCCL all-reduce across ranks — SURVEY.md §2.8 P5).
//
// Environment spec (must match the eager reference in
// evotorch_amd/neuroevolution/synthetic_env.py::rollout_eager, which
// quantizes the same operands to bf16):
//   obs_n = bf16((obs − mean) · inv_std)
//   a     = clip(W·obs_n + b, −1, 1)        (W bf16, accum fp32)
//   h     = bf16(V·obs)
//   o'    = tanh(Σ_i U_T[i]·h[i] + Σ_m D2_T[m]·a[m] + c)
//   r     = wr·o' + alive_bonus − act_cost·‖a‖²/A
//   o₀    = 0.1·philox_normal(member);  fitness = Σ_t r_t

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "philox.h"
#include "reduce.h"

namespace ea {

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a ROCm tensor")

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

__device__ __forceinline__ __bf16 f2b(float v) { return (__bf16)v; }
__device__ __forceinline__ float b2f(__bf16 v) { return (float)v; }

struct RolloutArgs {
    const float* params;      // [n_members][A*O + A]  (W row-major, then b)
    const float* env_blob;    // packed fp32 env data (see env_blob())
    float* fitness_out;       // [n_members]
    float* obs_stats_out;     // [2][O]  (sum, sumsq) — atomically accumulated
    int n_members;
    long member_offset;       // global id of member 0 (rank sharding)
    int obs_dim, act_dim, rank, steps;
    float alive_bonus, act_cost;
    unsigned long long init_seed;
};

// env_blob layout (fp32): V [R][O] · U_T [R][O] · D2_T [A][O] · c [O] ·
// wr [O] · mean [O] · std [O]

template <int kGroup>
__global__ __launch_bounds__(512, 2) void rollout_linear_kernel(RolloutArgs args) {
    const int O = args.obs_dim, A = args.act_dim, R = args.rank;
    const int tid = threadIdx.x;
    const int member = blockIdx.x;
    if (member >= args.n_members) return;
    const int A_PAD = (A + 2) & ~1;  // D2 per-output row stride (18 for A=17)
    const int R_PAD = (R + 2) & ~1;  // U per-output row stride

    extern __shared__ unsigned char lds_raw[];
    __bf16* W_l = reinterpret_cast<__bf16*>(lds_raw);   // [A][O] row-major
    __bf16* V_l = W_l + A * O;                          // [R][O] row-major
    // U and D2 are stored PAIR-INTERLEAVED COLUMN-MAJOR: element [i2][j]
    // is the bf16x2 (U_T[2*i2][j], U_T[2*i2+1][j]). Phase 2's thread j then
    // reads one 4 B bf16x2 per dot2 at consecutive-lane-consecutive-word
    // addresses — conflict-free b32 LDS reads. (The previous padded
    // per-output-row layout let the compiler merge the 9 contiguous pair
    // reads into strided ds_read_b128s: 1.16e9 bank-conflict cycles per
    // bench run, 58% of all LDS instructions.)
    bf16x2* U_pair = reinterpret_cast<bf16x2*>(V_l + R * O);  // [R_PAD/2][O]
    bf16x2* D2_pair = U_pair + (R_PAD / 2) * O;               // [A_PAD/2][O]
    __bf16* obs_b = reinterpret_cast<__bf16*>(D2_pair + (A_PAD / 2) * O);  // [O]
    __bf16* obsn_b = obs_b + O;                         // [O]    normalized
    __bf16* h_b = obsn_b + O;                           // [R_PAD]
    __bf16* act_b = h_b + R_PAD;                        // [A_PAD]
    float* c_l = reinterpret_cast<float*>(act_b + A_PAD);  // [O]
    float* wr_l = c_l + O;                              // [O]
    float* mean_l = wr_l + O;                           // [O]
    float* istd_l = mean_l + O;                         // [O]
    float* b_l = istd_l + O;                            // [A]
    float* scratch = b_l + A;                           // [8]

    // ---- stage ----
    const float* my_params = args.params + (long)member * (A * O + A);
    for (int i = tid; i < A * O; i += blockDim.x) W_l[i] = f2b(my_params[i]);
    for (int i = tid; i < A; i += blockDim.x) b_l[i] = my_params[A * O + i];
    {
        const float* e = args.env_blob;
        const int RO = R * O, AO = A * O;
        for (int i = tid; i < RO; i += blockDim.x) V_l[i] = f2b(e[i]);
        // U arrives as U_T [R][O]; store bf16x2 pairs along R, column-major
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
        // zero the h/act pads once so dot2 over padded vectors is exact
        for (int i = tid; i < R_PAD; i += blockDim.x) h_b[i] = f2b(0.0f);
        for (int i = tid; i < A_PAD; i += blockDim.x) act_b[i] = f2b(0.0f);
    }
    // initial observation: 0.1 * N(0,1), deterministic per global member id
    const unsigned long long gmember = (unsigned long long)(args.member_offset + member);
    for (int j4 = tid; j4 * 4 < O; j4 += blockDim.x) {
        float z[4];
        philox_normal4(args.init_seed, (uint32_t)gmember, (uint64_t)j4, z);
#pragma unroll
        for (int u = 0; u < 4; ++u) {
            const int j = j4 * 4 + u;
            if (j < O) obs_b[j] = f2b(0.1f * z[u]);
        }
    }
    __syncthreads();

    const int lane = tid & (kWaveSize - 1);
    const int wave = tid / kWaveSize;
    const int nwaves = blockDim.x / kWaveSize;
    const int n_pairs = O / 2;          // O must be even
    const int n_outputs = A + R;        // group-reduced dots per step

    // phase-1 layout: 8-lane groups, one output per group per round — the
    // wave computes 8 dots concurrently with only a 3-level shuffle
    // reduction (a 64-lane tree would serialize 6 dependent shuffles per
    // output; measured 10:1 SQ_WAIT:SQ_BUSY). With 4 waves × 8 groups,
    // all 33 outputs finish in ceil(33/32) = 2 rounds.
    constexpr int kRounds = 2;  // supports up to 2 * threads/kGroup outputs
    const int group = lane / kGroup;
    const int glane = lane % kGroup;
    const int groups_per_block = (int)(blockDim.x / kGroup);  // threads/8

    const __bf16* my_row[kRounds];
    bool my_is_act[kRounds];
    bool my_valid[kRounds];
    int my_out[kRounds];
    const int groups_per_wave = kWaveSize / kGroup;
#pragma unroll
    for (int r = 0; r < kRounds; ++r) {
        const int out = (wave * groups_per_wave + group) + r * groups_per_block;
        my_valid[r] = out < n_outputs;
        my_out[r] = my_valid[r] ? out : 0;
        my_is_act[r] = my_out[r] < A;
        my_row[r] = my_is_act[r] ? (W_l + my_out[r] * O) : (V_l + (my_out[r] - A) * O);
    }

    float fit_part = 0.0f;
    float actsq_part = 0.0f;
    float stat_sum[2] = {0.0f, 0.0f}, stat_sumsq[2] = {0.0f, 0.0f};

    // initial normalization (later steps fuse it into phase 2's epilogue)
    for (int j = tid; j < O; j += blockDim.x) {
        obsn_b[j] = f2b((b2f(obs_b[j]) - mean_l[j]) * istd_l[j]);
    }
    __syncthreads();

    for (int t = 0; t < args.steps; ++t) {
        // phase 1: all of this wave's dots concurrently (policy rows read
        // obsn, dynamics-factor rows read obs; both vectors loaded once
        // per pair-column and selected per output)
        float acc[kRounds];
#pragma unroll
        for (int r = 0; r < kRounds; ++r) acc[r] = 0.0f;
#pragma unroll
        for (int r = 0; r < kRounds; ++r) {
            if (!my_valid[r]) continue;
            const __bf16* row = my_row[r];
            const __bf16* vec = my_is_act[r] ? obsn_b : obs_b;
            float a0 = 0.0f, a1 = 0.0f;  // two chains: halve the serial depth
            const int full_iters = n_pairs / kGroup;   // guard-free iterations
            const int tail = n_pairs % kGroup;
            for (int i = 0; i + 1 < full_iters; i += 2) {
                const int p0 = glane + i * kGroup;
                const int p1 = glane + (i + 1) * kGroup;
                a0 = __builtin_amdgcn_fdot2_f32_bf16(
                    *reinterpret_cast<const bf16x2*>(row + 2 * p0),
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p0), a0, false);
                a1 = __builtin_amdgcn_fdot2_f32_bf16(
                    *reinterpret_cast<const bf16x2*>(row + 2 * p1),
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p1), a1, false);
            }
            if (full_iters & 1) {
                const int p0 = glane + (full_iters - 1) * kGroup;
                a0 = __builtin_amdgcn_fdot2_f32_bf16(
                    *reinterpret_cast<const bf16x2*>(row + 2 * p0),
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p0), a0, false);
            }
            if (tail && glane < tail) {
                const int p0 = glane + full_iters * kGroup;
                a1 = __builtin_amdgcn_fdot2_f32_bf16(
                    *reinterpret_cast<const bf16x2*>(row + 2 * p0),
                    *reinterpret_cast<const bf16x2*>(vec + 2 * p0), a1, false);
            }
            acc[r] = a0 + a1;
        }
        // 3-level in-group reductions (both rounds' shuffles overlap)
#pragma unroll
        for (int offset = kGroup / 2; offset > 0; offset >>= 1) {
#pragma unroll
            for (int r = 0; r < kRounds; ++r) {
                acc[r] += __shfl_down(acc[r], offset, kWaveSize);
            }
        }
        if (glane == 0) {
#pragma unroll
            for (int r = 0; r < kRounds; ++r) {
                if (!my_valid[r]) continue;
                if (my_is_act[r]) {
                    float a = fminf(fmaxf(acc[r] + b_l[my_out[r]], -1.0f), 1.0f);
                    act_b[my_out[r]] = f2b(a);
                    actsq_part = fmaf(a, a, actsq_part);
                } else {
                    h_b[my_out[r] - A] = f2b(acc[r]);
                }
            }
        }
        __syncthreads();

        // phase 2: per-thread dynamics rows (padded-stride LDS reads) with
        // the NEXT step's normalization fused into the epilogue
        for (int j = tid; j < O; j += blockDim.x) {
            // two independent dot chains (U·h and D2·a) halve the serial
            // dot2 dependency depth; joined at the end. Each operand read
            // is one bf16x2 at lane-consecutive addresses (conflict-free).
            float uacc = c_l[j], dacc = 0.0f;
#pragma unroll
            for (int p = 0; p < R_PAD / 2; ++p) {
                uacc = __builtin_amdgcn_fdot2_f32_bf16(
                    U_pair[p * O + j],
                    *reinterpret_cast<const bf16x2*>(h_b + 2 * p), uacc, false);
            }
#pragma unroll
            for (int p = 0; p < A_PAD / 2; ++p) {
                dacc = __builtin_amdgcn_fdot2_f32_bf16(
                    D2_pair[p * O + j],
                    *reinterpret_cast<const bf16x2*>(act_b + 2 * p), dacc, false);
            }
            const float o_new = tanhf(uacc + dacc);
            fit_part = fmaf(wr_l[j], o_new, fit_part);
            const int slot = j >= (int)blockDim.x;
            stat_sum[slot] += o_new;
            stat_sumsq[slot] = fmaf(o_new, o_new, stat_sumsq[slot]);
            const __bf16 ob = f2b(o_new);
            obs_b[j] = ob;
            obsn_b[j] = f2b((b2f(ob) - mean_l[j]) * istd_l[j]);
        }
        __syncthreads();
    }

    // ---- wrap-up: fitness reduction + obs-stat atomics ----
    float total = block_reduce_sum<false>(fit_part, scratch);
    __syncthreads();
    float act_total = block_reduce_sum<false>(actsq_part, scratch);
    if (tid == 0) {
        args.fitness_out[member] =
            total + args.alive_bonus * args.steps - args.act_cost * act_total / (float)A;
    }
    for (int j = tid; j < O; j += blockDim.x) {
        const int slot = j >= (int)blockDim.x;
        atomicAdd(&args.obs_stats_out[j], stat_sum[slot]);
        atomicAdd(&args.obs_stats_out[O + j], stat_sumsq[slot]);
    }
}

torch::Tensor rollout_linear(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out,
                             int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus,
                             double act_cost, int64_t init_seed, int64_t member_offset) {
    CHECK_GPU(params);
    TORCH_CHECK(params.is_contiguous() && params.dim() == 2, "params must be contiguous [N][L]");
    TORCH_CHECK(params.scalar_type() == at::ScalarType::Float, "params must be fp32");
    const int n = (int)params.size(0);
    const int O = (int)obs_dim, A = (int)act_dim, R = (int)rank;
    TORCH_CHECK(params.size(1) == (int64_t)A * O + A, "param length mismatch");
    TORCH_CHECK(O % 2 == 0, "obs_dim must be even (bf16x2 packing)");
    TORCH_CHECK(O <= 512, "obs_dim too large for the 2-slot stat accumulators");
    auto fitness = torch::empty({n}, params.options());

    RolloutArgs args;
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

    const int A_PAD = (A + 2) & ~1, R_PAD = (R + 2) & ~1;
    const size_t bf_elems = (size_t)A * O + (size_t)R * O + (size_t)O * R_PAD + (size_t)O * A_PAD +
                            2 * (size_t)O + R_PAD + A_PAD;
    const size_t f32_elems = 4 * (size_t)O + (size_t)A + 8;
    const size_t lds_bytes = bf_elems * 2 + f32_elems * 4 + 64;
    TORCH_CHECK(lds_bytes <= 64 * 1024, "rollout LDS footprint too large: ", lds_bytes,
                " bytes (reduce rank / dims)");
    auto stream = at::cuda::getCurrentCUDAStream();
    int block = 512;
    if (const char* env = getenv("EVOTORCH_AMD_ROLLOUT_BLOCK")) block = atoi(env);
    TORCH_CHECK(block == 256 || block == 512, "EVOTORCH_AMD_ROLLOUT_BLOCK must be 256 or 512");
    int group = 8;  // A/B-measured faster than 16 at O=376 (14.2 vs 15.5 us/step)
    if (const char* env = getenv("EVOTORCH_AMD_ROLLOUT_GROUP")) group = atoi(env);
    if (group == 8) {
        hipLaunchKernelGGL((rollout_linear_kernel<8>), dim3(n), dim3(block), lds_bytes, stream, args);
    } else {
        hipLaunchKernelGGL((rollout_linear_kernel<16>), dim3(n), dim3(block), lds_bytes, stream, args);
    }
    return fitness;
}

}  // namespace ea
