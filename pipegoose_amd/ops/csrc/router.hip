// Fused Switch-Transformer router for gfx950 (north-star kernel list).
//
// One pass over the [N, E] gate logits produces everything the MoE layer
// needs: per-token top-k expert ids + routing probs, the per-expert
// probability column sums (Switch aux-loss term P_i), per-expert top-1
// counts (f_i), and per-token logsumexp (ST-MoE z-loss) — instead of the
// softmax → topk → scatter → mean → logsumexp chain of separate kernels.
// E <= 64 (one expert per lane); the inference/dispatch path of the router
// (the training path keeps torch autograd ops for the loss gradients).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

template <typename T, int K>
__global__ void router_topk_kernel(
    const T* __restrict__ logits,
    int* __restrict__ topk_idx, float* __restrict__ topk_val,
    float* __restrict__ prob_colsum, int* __restrict__ top1_count,
    float* __restrict__ lse, int64_t N, int E) {
    // one wave per token row; lane e holds expert e
    const int64_t row = (int64_t)blockIdx.x * (blockDim.x / WAVE_SIZE)
                      + threadIdx.x / WAVE_SIZE;
    if (row >= N) return;
    const int lane = threadIdx.x % WAVE_SIZE;

    float x = (lane < E) ? to_float(logits[row * E + lane]) : -1e30f;
    float mx = wave_reduce_max(x);
    mx = __shfl(mx, 0, WAVE_SIZE);
    float ex = (lane < E) ? __expf(x - mx) : 0.f;
    float denom = wave_reduce_sum(ex);
    denom = __shfl(denom, 0, WAVE_SIZE);
    const float p = ex / denom;

    if (lane < E) atomicAdd(&prob_colsum[lane], p);
    if (lane == 0) lse[row] = mx + __logf(denom);

    // top-k by repeated masked max (K is 1 or 2)
    float pv = p;
#pragma unroll
    for (int kk = 0; kk < K; ++kk) {
        float best = wave_reduce_max(pv);
        best = __shfl(best, 0, WAVE_SIZE);
        // lowest lane holding the max wins
        const bool is_best = (pv == best) && (lane < E);
        unsigned long long mask = __ballot(is_best);
        const int win = __ffsll((unsigned long long)mask) - 1;
        if (lane == win) {
            topk_idx[row * K + kk] = lane;
            topk_val[row * K + kk] = p;
            if (kk == 0) atomicAdd(&top1_count[lane], 1);
            pv = -1.f;  // exclude from the next round
        }
    }
}

}  // namespace

std::vector<torch::Tensor> router_topk(torch::Tensor logits, int64_t k) {
    TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2);
    TORCH_CHECK(k == 1 || k == 2, "router_topk: k must be 1 or 2");
    const int64_t N = logits.size(0);
    const int E = logits.size(1);
    TORCH_CHECK(E <= 64, "router_topk: at most 64 experts (one per lane)");

    auto opts_f = logits.options().dtype(torch::kFloat);
    auto opts_i = logits.options().dtype(torch::kInt);
    auto topk_idx = torch::empty({N, k}, opts_i);
    auto topk_val = torch::empty({N, k}, opts_f);
    auto colsum = torch::zeros({E}, opts_f);
    auto count = torch::zeros({E}, opts_i);
    auto lse = torch::empty({N}, opts_f);

    constexpr int BLOCK = 256;
    const int wpb = BLOCK / WAVE_SIZE;
    dim3 grid((N + wpb - 1) / wpb);
    auto stream = at::cuda::getCurrentCUDAStream();

#define LAUNCH_RT(T, KV)                                                      \
    hipLaunchKernelGGL((router_topk_kernel<T, KV>), grid, dim3(BLOCK), 0,     \
        stream, reinterpret_cast<const T*>(logits.data_ptr()),                \
        topk_idx.data_ptr<int>(), topk_val.data_ptr<float>(),                 \
        colsum.data_ptr<float>(), count.data_ptr<int>(),                      \
        lse.data_ptr<float>(), N, E)

    if (logits.scalar_type() == torch::kFloat) {
        if (k == 1) LAUNCH_RT(float, 1); else LAUNCH_RT(float, 2);
    } else if (logits.scalar_type() == torch::kBFloat16) {
        if (k == 1) LAUNCH_RT(__hip_bfloat16, 1);
        else LAUNCH_RT(__hip_bfloat16, 2);
    } else {
        TORCH_CHECK(false, "router_topk: unsupported dtype");
    }
#undef LAUNCH_RT
    HIP_CHECK_LAUNCH();
    return {topk_idx, topk_val, colsum, count, lse};
}
