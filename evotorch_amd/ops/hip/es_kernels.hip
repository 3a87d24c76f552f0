// K1 / K3 / K4 kernels: Gaussian population sampling,
// fused ES gradient reductions, fused optimizer steps — gfx950 (CDNA4).
//
// Design notes (MI355X):
// * Sampling is HBM-write-bound: each thread produces 4 normals from one
//   philox counter and stores them contiguously (16 B/lane stores, the
//   coalescing sweet spot). The antithetic mirror writes the second half
//   of the population in the same pass — one kernel, one read of mu/sigma
//   (which L2-caches), ~N*L*4 bytes written.
// * Gradients are HBM-read-bound (read X once): grid = column tiles ×
//   row chunks; each block reduces its row chunk for 256 consecutive
//   columns (fully coalesced across the wave) and atomically adds its
//   fp32 partials — the fused kernel produces BOTH mu_grad and sigma_grad
//   from the single pass over X, halving traffic vs two library GEMVs.
// * ClipUp is three tiny kernels chained on the stream (norm, update+norm,
//   scale) — no host synchronization anywhere.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <algorithm>

#include "philox.h"
#include "reduce.h"

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a ROCm tensor")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

namespace ea {

// ---------------------------------------------------------------------------
// K1: sampling
// ---------------------------------------------------------------------------

// Graph-safe seeding: when `seed_ptr` is non-null the philox seed is read
// from device memory (and a separate bump kernel advances it), so a
// hipGraph replay of the sampling kernel produces a fresh population each
// replay — a by-value seed would be frozen into the captured graph.
__device__ __forceinline__ uint64_t resolve_seed(uint64_t seed, const unsigned long long* seed_ptr) {
    return seed_ptr ? (uint64_t)*seed_ptr : seed;
}

__global__ void bump_seed_kernel(unsigned long long* seed_ptr) {
    // splitmix64 advance (single thread)
    unsigned long long x = *seed_ptr + 0x9E3779B97F4A7C15ull;
    unsigned long long z = x;
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    *seed_ptr = (z ^ (z >> 31)) & 0x7FFFFFFFFFFFFFFFull;
}

// Counter addressing is STREAM-PER-ROW: row r of the (virtual, possibly
// rank-sharded) population draws from philox stream `row_offset + r` with
// the counter walking the row's columns (counter c covers columns
// [4c, 4c+4)). Any row partition of the population is therefore
// regenerable exactly, for ANY solution length — the property the SPMD
// sharded sampling and the streaming large-L gradient path both rely on
// (SURVEY.md §7 "RNG discipline"). CPU reference:
// evotorch_amd/neuroevolution/philox_ref.py::philox_normals_2d.
template <typename T, bool kSymmetric>
__global__ void sample_gaussian_kernel(T* __restrict__ out, const T* __restrict__ mu, const T* __restrict__ sigma,
                                       int64_t rows,  // = N (plain) or N/2 (symmetric)
                                       int64_t length, uint64_t seed_in,
                                       const unsigned long long* __restrict__ seed_ptr,
                                       int64_t row_offset) {
    const uint64_t seed = resolve_seed(seed_in, seed_ptr);
    const int64_t len4 = (length + 3) / 4;
    const int64_t total4 = rows * len4;
    for (int64_t idx4 = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx4 < total4;
         idx4 += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = idx4 / len4;
        const int64_t col4 = idx4 - row * len4;
        float z[4];
        philox_normal4(seed, (uint32_t)(row + row_offset), (uint64_t)col4, z);
        const int64_t base_col = col4 * 4;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int64_t col = base_col + j;
            if (col >= length) break;
            const float m = static_cast<float>(mu[col]);
            const float s = static_cast<float>(sigma[col]);
            const float plus = fmaf(s, z[j], m);
            out[row * length + col] = static_cast<T>(plus);
            if (kSymmetric) {
                out[(row + rows) * length + col] = static_cast<T>(2.0f * m - plus);
            }
        }
    }
}

// fp32 fast path for length % 4 == 0: the 4 philox normals of one counter
// are 4 consecutive elements of one row, stored as a single 16 B float4
// (the CDNA coalescing sweet spot — cdna_hip_programming.md G13); mu and
// sigma are read as float4 too.
template <bool kSymmetric>
__global__ void sample_gaussian_f32x4_kernel(float4* __restrict__ out, const float4* __restrict__ mu,
                                             const float4* __restrict__ sigma, int64_t rows, int64_t length4,
                                             uint64_t seed_in, const unsigned long long* __restrict__ seed_ptr,
                                             int64_t row_offset) {
    const uint64_t seed = resolve_seed(seed_in, seed_ptr);
    const int64_t total4 = rows * length4;
    for (int64_t idx4 = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx4 < total4;
         idx4 += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = idx4 / length4;
        const int64_t col4 = idx4 - row * length4;
        float z[4];
        philox_normal4(seed, (uint32_t)(row + row_offset), (uint64_t)col4, z);
        const float4 m = mu[col4];
        const float4 s = sigma[col4];
        float4 plus;
        plus.x = fmaf(s.x, z[0], m.x);
        plus.y = fmaf(s.y, z[1], m.y);
        plus.z = fmaf(s.z, z[2], m.z);
        plus.w = fmaf(s.w, z[3], m.w);
        out[idx4] = plus;
        if (kSymmetric) {
            float4 minus;
            minus.x = 2.0f * m.x - plus.x;
            minus.y = 2.0f * m.y - plus.y;
            minus.z = 2.0f * m.z - plus.z;
            minus.w = 2.0f * m.w - plus.w;
            out[idx4 + total4] = minus;
        }
    }
}

void sample_gaussian_impl(torch::Tensor out, torch::Tensor mu, torch::Tensor sigma, bool symmetric, int64_t seed,
                          const unsigned long long* seed_ptr, int64_t row_offset = 0) {
    TORCH_CHECK(row_offset >= 0, "row_offset must be non-negative");
    CHECK_GPU(out); CHECK_CONTIG(out); CHECK_GPU(mu); CHECK_GPU(sigma);
    const int64_t n = out.size(0), length = out.size(1);
    TORCH_CHECK(!symmetric || n % 2 == 0, "symmetric sampling needs even popsize");
    const int64_t rows = symmetric ? n / 2 : n;
    const int threads = 256;
    const int64_t len4 = (length + 3) / 4;
    const int64_t total4 = rows * len4;
    // 2048 blocks (8/CU) saturate: deeper grids measured identical (the
    // SQ_WAIT:BUSY ~12:1 is philox->Box-Muller dependency latency plus the
    // write pipe, not occupancy starvation)
    const int blocks = (int)std::min<int64_t>((total4 + threads - 1) / threads, 256 * 8);
    auto stream = at::cuda::getCurrentCUDAStream();
    if (out.scalar_type() == at::ScalarType::Float && length % 4 == 0 && mu.is_contiguous() && sigma.is_contiguous()) {
        if (symmetric) {
            hipLaunchKernelGGL((sample_gaussian_f32x4_kernel<true>), dim3(blocks), dim3(threads), 0, stream,
                               reinterpret_cast<float4*>(out.data_ptr<float>()), reinterpret_cast<const float4*>(mu.data_ptr<float>()),
                               reinterpret_cast<const float4*>(sigma.data_ptr<float>()), rows, length / 4, (uint64_t)seed, seed_ptr, row_offset);
        } else {
            hipLaunchKernelGGL((sample_gaussian_f32x4_kernel<false>), dim3(blocks), dim3(threads), 0, stream,
                               reinterpret_cast<float4*>(out.data_ptr<float>()), reinterpret_cast<const float4*>(mu.data_ptr<float>()),
                               reinterpret_cast<const float4*>(sigma.data_ptr<float>()), rows, length / 4, (uint64_t)seed, seed_ptr, row_offset);
        }
        return;
    }
    AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, out.scalar_type(), "sample_gaussian", [&] {
        using T = scalar_t;
        if (symmetric) {
            hipLaunchKernelGGL((sample_gaussian_kernel<T, true>), dim3(blocks), dim3(threads), 0, stream,
                               out.data_ptr<T>(), mu.data_ptr<T>(), sigma.data_ptr<T>(), rows, length, (uint64_t)seed, seed_ptr, row_offset);
        } else {
            hipLaunchKernelGGL((sample_gaussian_kernel<T, false>), dim3(blocks), dim3(threads), 0, stream,
                               out.data_ptr<T>(), mu.data_ptr<T>(), sigma.data_ptr<T>(), rows, length, (uint64_t)seed, seed_ptr, row_offset);
        }
    });
}

void sample_gaussian(torch::Tensor out, torch::Tensor mu, torch::Tensor sigma, bool symmetric, int64_t seed,
                     int64_t row_offset) {
    sample_gaussian_impl(out, mu, sigma, symmetric, seed, nullptr, row_offset);
}

// Graph-safe variant: the seed lives in `seed_buf` (int64 tensor of 1
// element) and is advanced ON DEVICE after sampling, so hipGraph replays
// draw fresh populations.
void sample_gaussian_graphsafe(torch::Tensor out, torch::Tensor mu, torch::Tensor sigma, bool symmetric,
                               torch::Tensor seed_buf) {
    TORCH_CHECK(seed_buf.is_cuda() && seed_buf.scalar_type() == at::ScalarType::Long && seed_buf.numel() >= 1,
                "seed_buf must be a cuda int64 tensor");
    auto* ptr = reinterpret_cast<unsigned long long*>(seed_buf.data_ptr<int64_t>());
    sample_gaussian_impl(out, mu, sigma, symmetric, 0, ptr);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(1), 0, stream, ptr);
}

// Advance a device seed chain by one splitmix64 step (standalone entry for
// graph-safe consumers like the rollout kernels: bump, then launch with
// seed_buf — replays draw fresh episode seeds).
void bump_seed(torch::Tensor seed_buf) {
    TORCH_CHECK(seed_buf.is_cuda() && seed_buf.scalar_type() == at::ScalarType::Long && seed_buf.numel() >= 1,
                "seed_buf must be a cuda int64 tensor");
    auto* ptr = reinterpret_cast<unsigned long long*>(seed_buf.data_ptr<int64_t>());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(1), 0, stream, ptr);
}

// Apply x = mu + sigma*z from PRE-GENERATED standard normals (the noise
// was filled on a side stream, hidden behind evaluation / collectives;
// only this cheap bandwidth-bound affine pass sits on the critical path
// after the distribution update — SURVEY.md §2.8 P2 overlap).
// Symmetric: z has N/2 rows; out rows [0,N/2) = mu+sigma*z, mirrored below.
template <typename T, bool kSymmetric>
__global__ void affine_from_noise_kernel(T* __restrict__ out, const float* __restrict__ z,
                                         const T* __restrict__ mu, const T* __restrict__ sigma,
                                         int64_t rows, int64_t length) {
    const int64_t total = rows * length;
    for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; e < total;
         e += (int64_t)gridDim.x * blockDim.x) {
        const int64_t col = e % length;
        const float m = static_cast<float>(mu[col]);
        const float s = static_cast<float>(sigma[col]);
        const float plus = fmaf(s, z[e], m);
        out[e] = static_cast<T>(plus);
        if (kSymmetric) out[e + total] = static_cast<T>(2.0f * m - plus);
    }
}

void affine_from_noise(torch::Tensor out, torch::Tensor z, torch::Tensor mu, torch::Tensor sigma, bool symmetric) {
    CHECK_GPU(out); CHECK_CONTIG(out); CHECK_GPU(z); CHECK_CONTIG(z);
    const int64_t n = out.size(0), length = out.size(1);
    TORCH_CHECK(!symmetric || n % 2 == 0, "symmetric affine needs even popsize");
    const int64_t rows = symmetric ? n / 2 : n;
    TORCH_CHECK(z.numel() == rows * length, "noise buffer shape mismatch");
    TORCH_CHECK(z.scalar_type() == at::ScalarType::Float, "noise must be fp32");
    const int threads = 256;
    const int blocks = (int)std::min<int64_t>((rows * length + threads - 1) / threads, 256 * 8);
    auto stream = at::cuda::getCurrentCUDAStream();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, out.scalar_type(), "affine_from_noise", [&] {
        using T = scalar_t;
        if (symmetric) {
            hipLaunchKernelGGL((affine_from_noise_kernel<T, true>), dim3(blocks), dim3(threads), 0, stream,
                               out.data_ptr<T>(), z.data_ptr<float>(), mu.data_ptr<T>(), sigma.data_ptr<T>(), rows, length);
        } else {
            hipLaunchKernelGGL((affine_from_noise_kernel<T, false>), dim3(blocks), dim3(threads), 0, stream,
                               out.data_ptr<T>(), z.data_ptr<float>(), mu.data_ptr<T>(), sigma.data_ptr<T>(), rows, length);
        }
    });
}

// ---------------------------------------------------------------------------
// K3: fused ES gradient reductions (N x L -> L), fp32 accumulation
// ---------------------------------------------------------------------------

enum class GradMode { kPlain, kSymmetric, kSnes };

template <typename T, GradMode kMode>
__global__ void es_grad_kernel(const T* __restrict__ x, const T* __restrict__ mu, const T* __restrict__ sigma,
                               const T* __restrict__ w, float* __restrict__ mu_grad, float* __restrict__ sigma_grad,
                               int64_t rows,  // = N (plain/snes) or N/2 directions (symmetric)
                               int64_t n_total, int64_t length, int row_chunks) {
    const int64_t col = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (col >= length) return;
    const int chunk = blockIdx.y;
    const int64_t r0 = (rows * chunk) / row_chunks;
    const int64_t r1 = (rows * (chunk + 1)) / row_chunks;
    const float m = static_cast<float>(mu[col]);
    const float s = static_cast<float>(sigma[col]);
    float acc_mu = 0.0f, acc_sigma = 0.0f;
    for (int64_t r = r0; r < r1; ++r) {
        const float xv = static_cast<float>(x[r * length + col]);
        const float noise = xv - m;
        if (kMode == GradMode::kSymmetric) {
            const float wp = static_cast<float>(w[r]);
            const float wm = static_cast<float>(w[r + rows]);
            acc_mu = fmaf(0.5f * (wp - wm), noise, acc_mu);
            acc_sigma = fmaf(0.5f * (wp + wm), (noise * noise - s * s) / s, acc_sigma);
        } else if (kMode == GradMode::kPlain) {
            const float wv = static_cast<float>(w[r]);
            acc_mu = fmaf(wv, noise, acc_mu);
            acc_sigma = fmaf(wv, (noise * noise - s * s) / s, acc_sigma);
        } else {  // SNES: raw-noise sigma gradient
            const float wv = static_cast<float>(w[r]);
            const float raw = noise / s;
            acc_mu = fmaf(wv, noise, acc_mu);
            acc_sigma = fmaf(wv, raw * raw - 1.0f, acc_sigma);
        }
    }
    // per-chunk partial rows (summed by a deterministic torch reduction) —
    // float atomicAdd would make the gradient run-to-run order-dependent
    mu_grad[(int64_t)chunk * length + col] = acc_mu;
    sigma_grad[(int64_t)chunk * length + col] = acc_sigma;
}

template <GradMode kMode>
std::vector<torch::Tensor> es_grad_launch(torch::Tensor samples, torch::Tensor mu, torch::Tensor sigma,
                                          torch::Tensor weights) {
    CHECK_GPU(samples); CHECK_CONTIG(samples);
    const int64_t n = samples.size(0), length = samples.size(1);
    const int64_t rows = (kMode == GradMode::kSymmetric) ? n / 2 : n;
    auto opts = samples.options().dtype(torch::kFloat32);
    const int threads = 256;
    const int col_blocks = (int)((length + threads - 1) / threads);
    // fill the chip: >= ~1024 blocks total
    int row_chunks = 1;
    if (col_blocks < 1024) row_chunks = (int)std::min<int64_t>((1024 + col_blocks - 1) / col_blocks, std::max<int64_t>(rows / 8, 1));
    auto mu_grad = torch::empty({row_chunks, length}, opts);
    auto sigma_grad = torch::empty({row_chunks, length}, opts);
    auto stream = at::cuda::getCurrentCUDAStream();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, samples.scalar_type(), "es_gradients", [&] {
        using T = scalar_t;
        hipLaunchKernelGGL((es_grad_kernel<T, kMode>), dim3(col_blocks, row_chunks), dim3(threads), 0, stream,
                           samples.data_ptr<T>(), mu.data_ptr<T>(), sigma.data_ptr<T>(), weights.data_ptr<T>(),
                           mu_grad.data_ptr<float>(), sigma_grad.data_ptr<float>(), rows, n, length, row_chunks);
    });
    if (row_chunks > 1) {
        return {mu_grad.sum(0).to(samples.scalar_type()), sigma_grad.sum(0).to(samples.scalar_type())};
    }
    return {mu_grad.reshape({length}).to(samples.scalar_type()), sigma_grad.reshape({length}).to(samples.scalar_type())};
}

std::vector<torch::Tensor> es_gradients(torch::Tensor samples, torch::Tensor mu, torch::Tensor sigma,
                                        torch::Tensor weights, bool symmetric) {
    if (symmetric) return es_grad_launch<GradMode::kSymmetric>(samples, mu, sigma, weights);
    return es_grad_launch<GradMode::kPlain>(samples, mu, sigma, weights);
}

std::vector<torch::Tensor> snes_gradients(torch::Tensor samples, torch::Tensor mu, torch::Tensor sigma,
                                          torch::Tensor weights) {
    return es_grad_launch<GradMode::kSnes>(samples, mu, sigma, weights);
}

// ---------------------------------------------------------------------------
// K4: ClipUp (3 chained kernels, no host sync) and Adam
// ---------------------------------------------------------------------------

template <typename T>
__global__ void sumsq_kernel(const T* __restrict__ v, float* __restrict__ partials, int64_t n) {
    __shared__ float scratch[8];
    float acc = 0.0f;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += (int64_t)gridDim.x * blockDim.x) {
        const float x = static_cast<float>(v[i]);
        acc = fmaf(x, x, acc);
    }
    acc = block_reduce_sum<false>(acc, scratch);
    if (threadIdx.x == 0) partials[blockIdx.x] = acc;
}

// Deterministic per-block partial combine (float atomicAdd would make
// ClipUp's norms — and hence the whole trajectory — run-to-run
// order-dependent): one block sums `count` partials in fixed order.
__global__ void reduce_partials_kernel(const float* __restrict__ partials, float* __restrict__ out, int count) {
    __shared__ float scratch[8];
    float acc = 0.0f;
    for (int i = threadIdx.x; i < count; i += blockDim.x) acc += partials[i];
    acc = block_reduce_sum<false>(acc, scratch);
    if (threadIdx.x == 0) *out = acc;
}

// velocity = momentum * velocity + grad * (step_size / ||grad||); also
// accumulates per-block ||velocity||^2 partials for the clip pass.
template <typename T>
__global__ void clipup_update_kernel(T* __restrict__ velocity, const T* __restrict__ grad,
                                     const float* __restrict__ gnorm_sq, float* __restrict__ vnorm_partials,
                                     float step_size, float momentum, int64_t n) {
    __shared__ float scratch[8];
    const float gnorm = sqrtf(fmaxf(*gnorm_sq, 1e-30f));
    const float scale = step_size / gnorm;
    float acc = 0.0f;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += (int64_t)gridDim.x * blockDim.x) {
        const float v = fmaf(momentum, static_cast<float>(velocity[i]), static_cast<float>(grad[i]) * scale);
        velocity[i] = static_cast<T>(v);
        acc = fmaf(v, v, acc);
    }
    acc = block_reduce_sum<false>(acc, scratch);
    if (threadIdx.x == 0) vnorm_partials[blockIdx.x] = acc;
}

template <typename T>
__global__ void clipup_clip_kernel(T* __restrict__ velocity, const float* __restrict__ vnorm_sq, float max_speed,
                                   int64_t n) {
    const float vnorm = sqrtf(fmaxf(*vnorm_sq, 1e-30f));
    const float scale = fminf(max_speed / vnorm, 1.0f);
    if (scale >= 1.0f) return;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += (int64_t)gridDim.x * blockDim.x) {
        velocity[i] = static_cast<T>(static_cast<float>(velocity[i]) * scale);
    }
}

void clipup_step(torch::Tensor velocity, torch::Tensor grad, double step_size, double max_speed, double momentum) {
    CHECK_GPU(velocity); CHECK_CONTIG(velocity);
    const int64_t n = velocity.numel();
    const int threads = 256;
    const int blocks = (int)std::min<int64_t>((n + threads - 1) / threads, 1024);
    // [0] gnorm², [1] vnorm², [2..2+blocks) per-block partials
    auto scratch = torch::empty({2 + blocks}, velocity.options().dtype(torch::kFloat32));
    auto stream = at::cuda::getCurrentCUDAStream();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, velocity.scalar_type(), "clipup_step", [&] {
        using T = scalar_t;
        float* gnorm_sq = scratch.data_ptr<float>();
        float* vnorm_sq = gnorm_sq + 1;
        float* partials = gnorm_sq + 2;
        hipLaunchKernelGGL((sumsq_kernel<T>), dim3(blocks), dim3(threads), 0, stream, grad.data_ptr<T>(), partials, n);
        hipLaunchKernelGGL(reduce_partials_kernel, dim3(1), dim3(256), 0, stream, partials, gnorm_sq, blocks);
        hipLaunchKernelGGL((clipup_update_kernel<T>), dim3(blocks), dim3(threads), 0, stream, velocity.data_ptr<T>(),
                           grad.data_ptr<T>(), gnorm_sq, partials, (float)step_size, (float)momentum, n);
        hipLaunchKernelGGL(reduce_partials_kernel, dim3(1), dim3(256), 0, stream, partials, vnorm_sq, blocks);
        hipLaunchKernelGGL((clipup_clip_kernel<T>), dim3(blocks), dim3(threads), 0, stream, velocity.data_ptr<T>(),
                           vnorm_sq, (float)max_speed, n);
    });
}

template <typename T>
__global__ void adam_kernel(T* __restrict__ step_out, const T* __restrict__ grad, T* __restrict__ m,
                            T* __restrict__ v, float stepsize, float beta1, float beta2, float epsilon,
                            float bias1, float bias2, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += (int64_t)gridDim.x * blockDim.x) {
        const float g = static_cast<float>(grad[i]);
        const float m_new = fmaf(beta1, static_cast<float>(m[i]), (1.0f - beta1) * g);
        const float v_new = fmaf(beta2, static_cast<float>(v[i]), (1.0f - beta2) * g * g);
        m[i] = static_cast<T>(m_new);
        v[i] = static_cast<T>(v_new);
        const float mhat = m_new / bias1;
        const float vhat = v_new / bias2;
        step_out[i] = static_cast<T>(stepsize * mhat / (sqrtf(vhat) + epsilon));
    }
}

// Graph-safe Adam: the step count lives in a device buffer and is
// advanced ON DEVICE, so hipGraph replays apply the correct bias
// correction each generation (a by-value step count would freeze it).
template <typename T>
__global__ void adam_graphsafe_kernel(T* __restrict__ step_out, const T* __restrict__ grad, T* __restrict__ m,
                                      T* __restrict__ v, const long long* __restrict__ t_buf, float stepsize,
                                      float beta1, float beta2, float epsilon, int64_t n) {
    const float t = (float)(*t_buf + 1);
    const float bias1 = 1.0f - powf(beta1, t);
    const float bias2 = 1.0f - powf(beta2, t);
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += (int64_t)gridDim.x * blockDim.x) {
        const float g = static_cast<float>(grad[i]);
        const float m_new = fmaf(beta1, static_cast<float>(m[i]), (1.0f - beta1) * g);
        const float v_new = fmaf(beta2, static_cast<float>(v[i]), (1.0f - beta2) * g * g);
        m[i] = static_cast<T>(m_new);
        v[i] = static_cast<T>(v_new);
        step_out[i] = static_cast<T>(stepsize * (m_new / bias1) / (sqrtf(v_new / bias2) + epsilon));
    }
}

__global__ void bump_step_kernel(long long* t_buf) { *t_buf += 1; }

void adam_step_graphsafe(torch::Tensor step_out, torch::Tensor grad, torch::Tensor m, torch::Tensor v,
                         torch::Tensor t_buf, double stepsize, double beta1, double beta2, double epsilon) {
    CHECK_GPU(step_out);
    TORCH_CHECK(t_buf.is_cuda() && t_buf.scalar_type() == at::ScalarType::Long && t_buf.numel() >= 1,
                "t_buf must be a cuda int64 tensor");
    const int64_t n = step_out.numel();
    const int threads = 256;
    const int blocks = (int)std::min<int64_t>((n + threads - 1) / threads, 2048);
    auto stream = at::cuda::getCurrentCUDAStream();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, step_out.scalar_type(), "adam_graphsafe", [&] {
        using T = scalar_t;
        hipLaunchKernelGGL((adam_graphsafe_kernel<T>), dim3(blocks), dim3(threads), 0, stream, step_out.data_ptr<T>(),
                           grad.data_ptr<T>(), m.data_ptr<T>(), v.data_ptr<T>(),
                           reinterpret_cast<const long long*>(t_buf.data_ptr<int64_t>()), (float)stepsize,
                           (float)beta1, (float)beta2, (float)epsilon, n);
    });
    hipLaunchKernelGGL(bump_step_kernel, dim3(1), dim3(1), 0, stream,
                       reinterpret_cast<long long*>(t_buf.data_ptr<int64_t>()));
}

void adam_step(torch::Tensor step_out, torch::Tensor grad, torch::Tensor m, torch::Tensor v, int64_t step_count,
               double stepsize, double beta1, double beta2, double epsilon) {
    CHECK_GPU(step_out);
    const int64_t n = step_out.numel();
    const int threads = 256;
    const int blocks = (int)std::min<int64_t>((n + threads - 1) / threads, 2048);
    const float bias1 = 1.0f - powf((float)beta1, (float)step_count);
    const float bias2 = 1.0f - powf((float)beta2, (float)step_count);
    auto stream = at::cuda::getCurrentCUDAStream();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, step_out.scalar_type(), "adam_step", [&] {
        using T = scalar_t;
        hipLaunchKernelGGL((adam_kernel<T>), dim3(blocks), dim3(threads), 0, stream, step_out.data_ptr<T>(),
                           grad.data_ptr<T>(), m.data_ptr<T>(), v.data_ptr<T>(), (float)stepsize, (float)beta1,
                           (float)beta2, (float)epsilon, bias1, bias2, n);
    });
}


// ---------------------------------------------------------------------------
// K2: fused fitness ranking -> utility map (reference
// tools/ranking.py:24-216). One launch replaces the torch chain
// (rocPRIM radix sort + scatter + 3-4 elementwise maps): a single-block
// bitonic sort over LDS (N <= 8192 padded to a power of two) followed by
// the utility map written back through the sorted index payload.
// method: 0 = centered (rank/(n-1) - 0.5), 1 = linear (rank/(n-1)),
//         2 = nes (log-utilities, normalized to sum to ~0).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(1024) void fused_rank_kernel(const float* __restrict__ fit, float* __restrict__ out,
                                                          int n, int p, int method, bool higher_better) {
    extern __shared__ unsigned char rank_lds[];
    float* keys = reinterpret_cast<float*>(rank_lds);
    int* idx = reinterpret_cast<int*>(keys + p);
    float* scratch = reinterpret_cast<float*>(idx + p);  // [32] for the nes sum

    const int tid = threadIdx.x;
    for (int i = tid; i < p; i += blockDim.x) {
        keys[i] = (i < n) ? (higher_better ? fit[i] : -fit[i]) : INFINITY;
        idx[i] = i;
    }
    __syncthreads();
    // bitonic sort ascending: position 0 = worst
    for (int k = 2; k <= p; k <<= 1) {
        for (int j = k >> 1; j > 0; j >>= 1) {
            for (int i = tid; i < p; i += blockDim.x) {
                const int ixj = i ^ j;
                if (ixj > i) {
                    const bool up = (i & k) == 0;
                    const float a = keys[i], b = keys[ixj];
                    // NaN-robust: order NaNs last so they rank as best-key
                    // (matches torch argsort's NaN-is-largest behavior)
                    const bool swap = up ? (b < a || (isnan(a) && !isnan(b)))
                                         : (a < b || (isnan(b) && !isnan(a)));
                    if (swap) {
                        keys[i] = b; keys[ixj] = a;
                        const int t = idx[i]; idx[i] = idx[ixj]; idx[ixj] = t;
                    }
                }
            }
            __syncthreads();
        }
    }
    if (method == 2) {
        // NES log-utilities need their global sum before the final map
        float partial = 0.0f;
        for (int i = tid; i < n; i += blockDim.x) {
            const float rank_from_best = (float)(n - i);
            const float u = fmaxf(__logf((float)n / 2.0f + 1.0f) - __logf(rank_from_best), 0.0f);
            keys[i] = u;  // reuse the key slot for the raw utility
            partial += u;
        }
        __syncthreads();
        // block sum (1024 threads -> 32 warp partials -> one value)
        const int lane = tid & 63, wave = tid >> 6;
        for (int off = 32; off > 0; off >>= 1) partial += __shfl_down(partial, off, 64);
        if (lane == 0) scratch[wave] = partial;
        __syncthreads();
        if (tid == 0) {
            float total = 0.0f;
            for (int w = 0; w < (int)(blockDim.x >> 6); ++w) total += scratch[w];
            scratch[0] = total;
        }
        __syncthreads();
        const float denom = scratch[0];
        const float inv_n = 1.0f / (float)n;
        for (int i = tid; i < n; i += blockDim.x) out[idx[i]] = keys[i] / denom - inv_n;
        return;
    }
    const float inv = (n > 1) ? 1.0f / (float)(n - 1) : 0.0f;
    for (int i = tid; i < n; i += blockDim.x) {
        const float r = (float)i;
        out[idx[i]] = (method == 0) ? (r * inv - 0.5f) : (r * inv);
    }
}

torch::Tensor fused_rank(torch::Tensor fitnesses, int64_t method, bool higher_better) {
    TORCH_CHECK(fitnesses.is_cuda() && fitnesses.dim() == 1, "fused_rank expects a 1-D ROCm tensor");
    auto fit = fitnesses.to(torch::kFloat32).contiguous();
    const int n = (int)fit.size(0);
    TORCH_CHECK(n >= 1 && n <= 8192, "fused_rank supports 1 <= n <= 8192");
    int p = 1;
    while (p < n) p <<= 1;
    auto out = torch::empty({n}, fit.options());
    const size_t lds = (size_t)p * 8 + 32 * 4;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(fused_rank_kernel, dim3(1), dim3(1024), lds, stream, fit.data_ptr<float>(),
                       out.data_ptr<float>(), n, p, (int)method, higher_better);
    return out;
}

}  // namespace ea
