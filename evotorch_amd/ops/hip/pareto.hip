// K7: NSGA-II non-dominated sorting — domination counts
// and iterative front peeling WITHOUT materializing the N×N domination
// matrix (the eager path's (N, N, M) broadcast explodes past ~10k
// solutions; these kernels are O(N²·M) compute but O(N) memory).
//
// utils is (N, M) fp32 with HIGHER IS BETTER for every column (senses
// already folded by the caller). Each thread owns one solution j, caches
// its own objective row in registers, and streams the other rows — row i
// is read by all threads of a wave simultaneously (broadcast, L2-cached).
//
// Peeling: counts[j] = #dominators. Front k = {count == 0}. The peel
// kernel subtracts each front member's contributions; every solution is
// subtracted exactly once over the whole sort, so total peel work is one
// more O(N²·M) pass — host sync only once per front.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace ea {

constexpr int kMaxObjCache = 16;

__device__ __forceinline__ bool dominates_rows(const float* __restrict__ a, const float* __restrict__ b, int m) {
    bool ge_all = true, gt_any = false;
    for (int k = 0; k < m; ++k) {
        ge_all &= (a[k] >= b[k]);
        gt_any |= (a[k] > b[k]);
    }
    return ge_all && gt_any;
}

__global__ void domination_counts_kernel(const float* __restrict__ utils, int* __restrict__ counts, int64_t n, int m) {
    const int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (j >= n) return;
    float mine[kMaxObjCache];
    const bool cached = m <= kMaxObjCache;
    if (cached) {
        for (int k = 0; k < m; ++k) mine[k] = utils[j * m + k];
    }
    int count = 0;
    for (int64_t i = 0; i < n; ++i) {
        const float* other = utils + i * m;
        bool ge_all = true, gt_any = false;
        if (cached) {
            for (int k = 0; k < m; ++k) {
                const float o = other[k];
                ge_all &= (o >= mine[k]);
                gt_any |= (o > mine[k]);
            }
        } else {
            const float* me = utils + j * m;
            for (int k = 0; k < m; ++k) {
                const float o = other[k];
                ge_all &= (o >= me[k]);
                gt_any |= (o > me[k]);
            }
        }
        count += (ge_all && gt_any) ? 1 : 0;
    }
    counts[j] = count;
}

// Front peeling without host syncs, in two device-side stages per front:
//   compact: zero-count unassigned solutions claim a slot in front_list
//            (atomic counter), get their rank assigned, and are marked.
//   subtract: every still-unassigned solution subtracts the domination
//            contributions of the compacted front members ONLY — total
//            subtract work over the whole sort is one O(N²·M) pass
//            (each solution appears in exactly one front), instead of the
//            O(N²·fronts) full-mask scans of a mask-based peel.

__global__ void compact_front_kernel(int* __restrict__ counts, int64_t* __restrict__ ranks,
                                     int* __restrict__ front_list, int* __restrict__ front_count, int64_t n,
                                     int64_t front_index) {
    const int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (j == 0 && front_index == 0) {
        // nothing: front_count reset is handled by reset_count_kernel
    }
    if (j >= n) return;
    if (counts[j] == 0) {
        const int pos = atomicAdd(front_count, 1);
        front_list[pos] = (int)j;
        ranks[j] = front_index;
        counts[j] = -1;  // assigned marker
    }
}

__global__ void reset_count_kernel(int* __restrict__ front_count) { *front_count = 0; }

__global__ void zero_counts_kernel(int* __restrict__ front_counts, int batch) {
    const int j = blockIdx.x * blockDim.x + threadIdx.x;
    if (j < batch) front_counts[j] = 0;
}

__global__ void subtract_front_kernel(const float* __restrict__ utils, int* __restrict__ counts,
                                      const int* __restrict__ front_list, const int* __restrict__ front_count,
                                      int64_t n, int m) {
    const int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (j >= n) return;
    if (counts[j] < 0) return;  // already assigned
    float mine[kMaxObjCache];
    const bool cached = m <= kMaxObjCache;
    if (cached) {
        for (int k = 0; k < m; ++k) mine[k] = utils[j * m + k];
    }
    const int fc = *front_count;
    int removed = 0;
    for (int t = 0; t < fc; ++t) {
        const float* other = utils + (int64_t)front_list[t] * m;
        const float* me = cached ? mine : (utils + j * m);
        bool ge_all = true, gt_any = false;
        for (int k = 0; k < m; ++k) {
            const float o = other[k];
            ge_all &= (o >= me[k]);
            gt_any |= (o > me[k]);
        }
        removed += (ge_all && gt_any) ? 1 : 0;
    }
    counts[j] -= removed;
}

torch::Tensor domination_counts(torch::Tensor utils) {
    TORCH_CHECK(utils.is_cuda() && utils.dim() == 2, "utils must be a 2-D ROCm tensor");
    auto utils_f = utils.to(torch::kFloat32).contiguous();
    const int64_t n = utils_f.size(0);
    const int m = (int)utils_f.size(1);
    auto counts = torch::empty({n}, utils_f.options().dtype(torch::kInt32));
    const int threads = 256;
    const int blocks = (int)((n + threads - 1) / threads);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(domination_counts_kernel, dim3(blocks), dim3(threads), 0, stream, utils_f.data_ptr<float>(),
                       counts.data_ptr<int>(), n, m);
    return counts;
}

torch::Tensor pareto_ranks(torch::Tensor utils, int64_t min_assigned) {
    // min_assigned > 0: stop peeling once that many solutions hold final
    // ranks (take_best(n) only needs the fronts crossing n); stragglers
    // get a beyond-last rank. 0 = full sort.
    TORCH_CHECK(utils.is_cuda() && utils.dim() == 2, "utils must be a 2-D ROCm tensor");
    auto utils_f = utils.to(torch::kFloat32).contiguous();
    const int64_t n = utils_f.size(0);
    const int m = (int)utils_f.size(1);
    if (min_assigned <= 0 || min_assigned > n) min_assigned = n;
    auto counts = domination_counts(utils_f);
    auto ranks = torch::full({n}, -1, utils_f.options().dtype(torch::kInt64));
    const int threads = 256;
    const int blocks = (int)((n + threads - 1) / threads);
    auto stream = at::cuda::getCurrentCUDAStream();
    // Peel in blind batches of kPeelBatch fronts with ONE host sync per
    // batch (a sync per front costs hundreds of round-trips at large N:
    // a random 16k population can have hundreds of fronts). A peel with
    // an empty front is a no-op, so over-issuing is safe.
    auto front_list = torch::empty({n}, utils_f.options().dtype(torch::kInt32));
    // per-front counter slots: zeroing them in ONE launch per batch (not
    // one reset launch per front) cuts the peel's kernel launches by a
    // third — at 32k population with ~350 fronts the loop is LAUNCH-bound
    constexpr int kPeelBatch = 16;
    auto front_counts = torch::zeros({kPeelBatch}, utils_f.options().dtype(torch::kInt32));
    int64_t front_index = 0;
    while (front_index <= n) {
        hipLaunchKernelGGL(zero_counts_kernel, dim3(1), dim3(kPeelBatch), 0, stream, front_counts.data_ptr<int>(),
                           kPeelBatch);
        for (int k = 0; k < kPeelBatch; ++k) {
            int* slot = front_counts.data_ptr<int>() + k;
            hipLaunchKernelGGL(compact_front_kernel, dim3(blocks), dim3(threads), 0, stream, counts.data_ptr<int>(),
                               ranks.data_ptr<int64_t>(), front_list.data_ptr<int>(), slot, n,
                               front_index);
            hipLaunchKernelGGL(subtract_front_kernel, dim3(blocks), dim3(threads), 0, stream, utils_f.data_ptr<float>(),
                               counts.data_ptr<int>(), front_list.data_ptr<int>(), slot, n, m);
            ++front_index;
        }
        const int64_t remaining = (counts >= 0).sum().item<int64_t>();  // one sync per batch
        if (remaining == 0) break;
        if (n - remaining >= min_assigned) {
            // enough solutions carry final ranks; lump the rest one past
            // the last peeled front (they sort after every real front)
            ranks.masked_fill_(counts >= 0, front_index);
            break;
        }
        // numerical corner: no zero-count candidates among the remaining —
        // assign the stragglers and stop
        const int64_t candidates = (counts == 0).sum().item<int64_t>();
        if (candidates == 0) {
            ranks.masked_fill_(counts >= 0, front_index);
            break;
        }
    }
    return ranks;
}

// ---------------------------------------------------------------------------
// K9: MAPElites cell assignment — for every hypergrid
// cell, the best solution whose features fall inside the cell box
// (reference mapelites.py:24-68). The eager path materializes a (C, N)
// inside-mask + masked utility matrix (O(C·N) memory — 800 MB at
// C=10k × N=20k); this kernel streams the solutions once per cell thread
// with O(C + N) memory. Feature rows are read wave-uniformly (broadcast,
// L2-cached) like the domination kernels above.
// ---------------------------------------------------------------------------

constexpr int kMaxFeat = 16;

__global__ void mapelites_assign_kernel(const float* __restrict__ grid,   // [C][F][2]
                                        const float* __restrict__ feats,  // [N][F]
                                        const float* __restrict__ utils,  // [N] higher is better
                                        int64_t* __restrict__ best_idx,   // [C]
                                        bool* __restrict__ any_valid,     // [C]
                                        int64_t c_total, int64_t n, int f) {
    const int64_t c = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (c >= c_total) return;
    float lo[kMaxFeat], hi[kMaxFeat];
    for (int k = 0; k < f; ++k) {
        lo[k] = grid[(c * f + k) * 2];
        hi[k] = grid[(c * f + k) * 2 + 1];
    }
    float best_u = -INFINITY;
    int64_t best = 0;
    bool found = false;
    for (int64_t i = 0; i < n; ++i) {
        bool inside = true;
        for (int k = 0; k < f; ++k) {
            const float x = feats[i * f + k];
            inside &= (x >= lo[k]) & (x <= hi[k]);
        }
        const float u = utils[i];
        // strict > keeps the FIRST best row on ties (torch argmax parity)
        if (inside && (!found || u > best_u)) {
            best_u = u;
            best = i;
            found = true;
        }
    }
    best_idx[c] = best;
    any_valid[c] = found;
}

std::vector<torch::Tensor> mapelites_assign(torch::Tensor grid, torch::Tensor feats, torch::Tensor utils) {
    TORCH_CHECK(grid.is_cuda() && grid.dim() == 3 && grid.size(2) == 2, "grid must be (C, F, 2) on ROCm");
    TORCH_CHECK(feats.dim() == 2 && utils.dim() == 1, "feats (N, F), utils (N)");
    const int f = (int)grid.size(1);
    TORCH_CHECK(f <= kMaxFeat, "at most ", kMaxFeat, " feature dimensions");
    auto grid_f = grid.to(torch::kFloat32).contiguous();
    auto feats_f = feats.to(torch::kFloat32).contiguous();
    auto utils_f = utils.to(torch::kFloat32).contiguous();
    const int64_t c_total = grid.size(0), n = feats.size(0);
    auto best = torch::zeros({c_total}, grid.options().dtype(torch::kInt64));
    auto valid = torch::zeros({c_total}, grid.options().dtype(torch::kBool));
    const int threads = 256;
    const int blocks = (int)((c_total + threads - 1) / threads);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(mapelites_assign_kernel, dim3(blocks), dim3(threads), 0, stream, grid_f.data_ptr<float>(),
                       feats_f.data_ptr<float>(), utils_f.data_ptr<float>(), best.data_ptr<int64_t>(),
                       valid.data_ptr<bool>(), c_total, n, f);
    return {best, valid};
}

}  // namespace ea
