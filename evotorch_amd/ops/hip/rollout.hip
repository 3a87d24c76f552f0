// K10+K11 fused: whole-generation rollout of a population of linear
// policies through the synthetic vectorized environment, one kernel launch
// per generation (SURVEY.md §3.4 — the VecGymNE hot loop, collapsed).
//
// MI355X design: one workgroup per population member. The member's policy
// weights AND the (shared) environment matrices are staged once into LDS as
// bf16 and stay resident for the entire T-step episode — the inner loop
// touches no global memory at all, so the rollout is pure VALU compute
// instead of the reference's per-step kernel round-trips through HBM
// (policy W re-read every step). Observation-normalization statistics
// (sum, sumsq) accumulate per-thread and are merged with one atomic pass
// at the end (K11; they become a single RCCL all-reduce across ranks —
// SURVEY.md §2.8 P5).
//
// Environment spec (must match the eager reference in
// evotorch_amd/neuroevolution/synthetic_env.py):
//   state  o ∈ R^O, action a = clip(W·obs_norm(o) + b, -1, 1) ∈ R^A
//   h  = V·o                       (V: R×O, low-rank dynamics factor)
//   o' = tanh(Uᵀ·h + D2ᵀ·a + c)    (U stored transposed [R][O], D2 [A][O])
//   r  = wr·o' + alive_bonus − act_cost·‖a‖²/A
//   fitness = Σ_t r_t over T steps; o_0 = 0.1·philox_normal(member)
// All matrices are bf16 in LDS; accumulation is fp32.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "philox.h"
#include "reduce.h"

namespace ea {

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a ROCm tensor")

__device__ __forceinline__ float bf2f(__hip_bfloat16 v) { return __bfloat162float(v); }
__device__ __forceinline__ __hip_bfloat16 f2bf(float v) { return __float2bfloat16(v); }

struct RolloutArgs {
    const float* params;      // [n_members][A*O + A]  (W row-major, then b)
    const float* env_blob;    // packed bf16/fp32 env data, see offsets below
    float* fitness_out;       // [n_members]
    float* obs_stats_out;     // [2][O]  (sum, sumsq) — atomically accumulated
    int n_members;
    long member_offset;       // global id of member 0 (rank sharding)
    int obs_dim, act_dim, rank, steps;
    float alive_bonus, act_cost;
    unsigned long long init_seed;
};

// env_blob layout (all fp32, converted to bf16 while staging into LDS):
//   V     [rank][O]
//   U_T   [rank][O]   (U transposed: U_T[i][j] = U[j][i])
//   D2_T  [act][O]
//   c     [O]
//   wr    [O]
//   mean  [O]
//   std   [O]         (already max(std, eps))

__global__ __launch_bounds__(256, 2) void rollout_linear_kernel(RolloutArgs args) {
    const int O = args.obs_dim, A = args.act_dim, R = args.rank;
    const int tid = threadIdx.x;
    const int member = blockIdx.x;
    if (member >= args.n_members) return;

    extern __shared__ unsigned char lds_raw[];
    __hip_bfloat16* W_l = reinterpret_cast<__hip_bfloat16*>(lds_raw);            // [A][O]
    __hip_bfloat16* V_l = W_l + A * O;                                           // [R][O]
    __hip_bfloat16* UT_l = V_l + R * O;                                          // [R][O]
    __hip_bfloat16* D2T_l = UT_l + R * O;                                        // [A][O]
    __hip_bfloat16* c_l = D2T_l + A * O;                                         // [O]
    __hip_bfloat16* wr_l = c_l + O;                                              // [O]
    __hip_bfloat16* mean_l = wr_l + O;                                           // [O]
    __hip_bfloat16* std_l = mean_l + O;                                          // [O]
    float* b_l = reinterpret_cast<float*>(std_l + O);                            // [A]
    float* obs = b_l + A;                                                        // [O]
    float* obs_n = obs + O;                                                      // [O]
    float* h_l = obs_n + O;                                                      // [R]
    float* act_l = h_l + R;                                                      // [A]
    float* scratch = act_l + A;                                                  // [8]

    // ---- stage: policy params (per member) + env matrices (shared) ----
    const float* my_params = args.params + (long)member * (A * O + A);
    for (int i = tid; i < A * O; i += blockDim.x) W_l[i] = f2bf(my_params[i]);
    for (int i = tid; i < A; i += blockDim.x) b_l[i] = my_params[A * O + i];
    {
        const float* e = args.env_blob;
        const int RO = R * O, AO = A * O;
        for (int i = tid; i < RO; i += blockDim.x) V_l[i] = f2bf(e[i]);
        for (int i = tid; i < RO; i += blockDim.x) UT_l[i] = f2bf(e[RO + i]);
        for (int i = tid; i < AO; i += blockDim.x) D2T_l[i] = f2bf(e[2 * RO + i]);
        const float* tail = e + 2 * RO + AO;
        for (int i = tid; i < O; i += blockDim.x) {
            c_l[i] = f2bf(tail[i]);
            wr_l[i] = f2bf(tail[O + i]);
            mean_l[i] = f2bf(tail[2 * O + i]);
            std_l[i] = f2bf(tail[3 * O + i]);
        }
    }
    // initial observation: 0.1 * N(0,1), deterministic per global member id
    const unsigned long long gmember = (unsigned long long)(args.member_offset + member);
    for (int j4 = tid; j4 * 4 < O; j4 += blockDim.x) {
        float z[4];
        philox_normal4(args.init_seed, (uint32_t)gmember, (uint64_t)j4, z);
#pragma unroll
        for (int u = 0; u < 4; ++u) {
            const int j = j4 * 4 + u;
            if (j < O) obs[j] = 0.1f * z[u];
        }
    }
    __syncthreads();

    const int lane = tid & (kWaveSize - 1);
    const int wave = tid / kWaveSize;
    const int nwaves = blockDim.x / kWaveSize;

    float fit_part = 0.0f;       // per-thread fitness partial (dims owned)
    float actsq_part = 0.0f;     // per-thread Σ_t a²   (acts owned by wave 0 lanes)
    float stat_sum[8], stat_sumsq[8];
    const int dims_per_thread = (O + blockDim.x - 1) / blockDim.x;
#pragma unroll
    for (int u = 0; u < 8; ++u) { stat_sum[u] = 0.0f; stat_sumsq[u] = 0.0f; }

    for (int t = 0; t < args.steps; ++t) {
        // phase 1: obs normalization (owned dims)
        for (int j = tid; j < O; j += blockDim.x) {
            obs_n[j] = (obs[j] - bf2f(mean_l[j])) / bf2f(std_l[j]);
        }
        __syncthreads();

        // phase 2: policy  a[j] = clip(W[j]·obs_n + b[j])  and  h[i] = V[i]·obs
        for (int j = wave; j < A; j += nwaves) {
            float acc = 0.0f;
            for (int k = lane; k < O; k += kWaveSize) acc = fmaf(bf2f(W_l[j * O + k]), obs_n[k], acc);
            acc = wave_reduce_sum(acc);
            if (lane == 0) {
                float a = acc + b_l[j];
                a = fminf(fmaxf(a, -1.0f), 1.0f);
                act_l[j] = a;
            }
        }
        for (int i = wave; i < R; i += nwaves) {
            float acc = 0.0f;
            for (int k = lane; k < O; k += kWaveSize) acc = fmaf(bf2f(V_l[i * O + k]), obs[k], acc);
            acc = wave_reduce_sum(acc);
            if (lane == 0) h_l[i] = acc;
        }
        __syncthreads();

        // action cost accumulation (wave 0, lanes over A)
        if (wave == 0 && lane < A) {
            const float a = act_l[lane];
            actsq_part = fmaf(a, a, actsq_part);
        }

        // phase 3: dynamics  o'[j] = tanh(c[j] + Σ_i U_T[i][j] h[i] + Σ_m D2_T[m][j] a[m])
        for (int j = tid; j < O; j += blockDim.x) {
            float v = bf2f(c_l[j]);
            for (int i = 0; i < R; ++i) v = fmaf(bf2f(UT_l[i * O + j]), h_l[i], v);
            for (int m = 0; m < A; ++m) v = fmaf(bf2f(D2T_l[m * O + j]), act_l[m], v);
            const float o_new = tanhf(v);
            fit_part = fmaf(bf2f(wr_l[j]), o_new, fit_part);
            const int slot = (j - tid) / (int)blockDim.x;
            stat_sum[slot] += o_new;
            stat_sumsq[slot] = fmaf(o_new, o_new, stat_sumsq[slot]);
            obs[j] = o_new;  // in-place: raw obs of step t is dead after phase 2
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
        const int slot = (j - tid) / (int)blockDim.x;
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

    const size_t bf_elems = (size_t)(A * O) + 2 * (size_t)(R * O) + (size_t)(A * O) + 4 * (size_t)O;
    const size_t f32_elems = (size_t)A + 2 * (size_t)O + (size_t)R + (size_t)A + 8;
    const size_t lds_bytes = bf_elems * 2 + f32_elems * 4 + 64;
    TORCH_CHECK(lds_bytes <= 64 * 1024, "rollout LDS footprint too large: ", lds_bytes,
                " bytes (reduce rank / dims)");
    TORCH_CHECK((O + 255) / 256 <= 8, "obs_dim too large for stat slots");
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(rollout_linear_kernel, dim3(n), dim3(256), lds_bytes, stream, args);
    return fitness;
}

}  // namespace ea
