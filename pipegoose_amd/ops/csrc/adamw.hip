// Fused multi-tensor AdamW for gfx950 — ONE kernel per optimizer step.
//
// torch's foreach AdamW walks the tensor list with several elementwise
// kernels per step (mul, add, addcmul, sqrt, div ...), re-reading exp_avg /
// exp_avg_sq from HBM each pass; this kernel does the whole decoupled-AdamW
// update in a single pass: read p,g,m,v -> write p,m,v (minimum possible
// HBM traffic for torch-compatible per-param fp32 state).
//
// Tensor list handling: the host packs pointers + a slab table (one entry
// per 64K-element slab: tensor index, slab index) into one small device
// buffer; each workgroup serves one slab.  State layout stays one
// exp_avg/exp_avg_sq tensor per param, so state_dict round-trips with
// torch.optim.AdamW ("bitwise-compatible state").
//
// No reference counterpart (reference used plain torch.optim over ZeRO-1
// shards, optim/zero/optim.py:57-66); SURVEY §2.7 item 1 "grad-scale /
// flatten utilities for ZeRO" is subsumed by updating shards in place.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

constexpr long SLAB = 65536;  // elements per workgroup

template <typename T, typename TS>
__global__ __launch_bounds__(256)
void adamw_chunked_kernel(const unsigned long long* __restrict__ ptrs,
                          const int* __restrict__ slab_tensor,
                          const int* __restrict__ slab_idx,
                          const long* __restrict__ numels,
                          int n_tensors,
                          float lr, float beta1, float beta2, float eps,
                          float wd_factor,   // 1 - lr * weight_decay
                          float step_size,   // lr / (1 - beta1^t)
                          float bc2_rsqrt) { // 1 / sqrt(1 - beta2^t)
    const int s = blockIdx.x;
    const int t = slab_tensor[s];
    T* p = reinterpret_cast<T*>(ptrs[t]);
    const T* g = reinterpret_cast<const T*>(ptrs[n_tensors + t]);
    TS* m = reinterpret_cast<TS*>(ptrs[2 * n_tensors + t]);
    TS* v = reinterpret_cast<TS*>(ptrs[3 * n_tensors + t]);
    const long n = numels[t];
    const long base = (long)slab_idx[s] * SLAB;
    const long end = min(base + SLAB, n);

    const float c1 = 1.0f - beta1, c2 = 1.0f - beta2;
    // vectorized main body: 8 elements per lane per iteration
    long i = base + (long)threadIdx.x * 8;
    for (; i + 8 <= end; i += (long)blockDim.x * 8) {
        typename vec8<T>::type pv8 = *reinterpret_cast<const typename vec8<T>::type*>(p + i);
        typename vec8<T>::type gv8 = *reinterpret_cast<const typename vec8<T>::type*>(g + i);
        typename vec8<TS>::type mv = *reinterpret_cast<const typename vec8<TS>::type*>(m + i);
        typename vec8<TS>::type vv = *reinterpret_cast<const typename vec8<TS>::type*>(v + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const float gj = to_float(gv8.v[j]);
            const float mj = beta1 * to_float(mv.v[j]) + c1 * gj;
            const float vj = beta2 * to_float(vv.v[j]) + c2 * gj * gj;
            mv.v[j] = from_float<TS>(mj);
            vv.v[j] = from_float<TS>(vj);
            const float denom = sqrtf(vj) * bc2_rsqrt + eps;
            pv8.v[j] = from_float<T>(
                to_float(pv8.v[j]) * wd_factor - step_size * mj / denom);
        }
        *reinterpret_cast<typename vec8<T>::type*>(p + i) = pv8;
        *reinterpret_cast<typename vec8<TS>::type*>(m + i) = mv;
        *reinterpret_cast<typename vec8<TS>::type*>(v + i) = vv;
    }
    // handle the ragged tail with the whole block, element-wise
    const long vec_end = base + ((end - base) / 8) * 8;
    for (long k = vec_end + threadIdx.x; k < end; k += blockDim.x) {
        const float gj = to_float(g[k]);
        const float mj = beta1 * to_float(m[k]) + c1 * gj;
        const float vj = beta2 * to_float(v[k]) + c2 * gj * gj;
        m[k] = from_float<TS>(mj);
        v[k] = from_float<TS>(vj);
        const float denom = sqrtf(vj) * bc2_rsqrt + eps;
        p[k] = from_float<T>(to_float(p[k]) * wd_factor - step_size * mj / denom);
    }
}

}  // namespace

void adamw_fused_step(std::vector<torch::Tensor> params,
                      std::vector<torch::Tensor> grads,
                      std::vector<torch::Tensor> exp_avgs,
                      std::vector<torch::Tensor> exp_avg_sqs,
                      torch::Tensor meta_dev,  // packed ptrs + slab table
                      int64_t n_slabs,
                      double lr, double beta1, double beta2, double eps,
                      double weight_decay, int64_t step) {
    TORCH_CHECK(!params.empty());
    const int T_ = (int)params.size();
    const auto dt = params[0].scalar_type();
    const auto st_dt = exp_avgs[0].scalar_type();
    const float bc1 = 1.0f - powf((float)beta1, (float)step);
    const float bc2 = 1.0f - powf((float)beta2, (float)step);
    const float step_size = (float)lr / bc1;
    const float bc2_rsqrt = 1.0f / sqrtf(bc2);
    const float wd_factor = 1.0f - (float)(lr * weight_decay);

    // meta_dev layout (see fused_adamw.py): [4*T u64 ptrs][T i64 numels]
    // [n_slabs i32 tensor][n_slabs i32 slab]
    auto base = meta_dev.data_ptr<uint8_t>();
    auto ptrs = reinterpret_cast<const unsigned long long*>(base);
    auto numels = reinterpret_cast<const long*>(base + 4 * T_ * 8);
    auto slab_tensor = reinterpret_cast<const int*>(base + 5 * T_ * 8);
    auto slab_idx = slab_tensor + n_slabs;

    auto stream = at::cuda::getCurrentCUDAStream();
    dim3 grid((unsigned)n_slabs);
#define LAUNCH_ADAMW(TP, TSP)                                                 \
    hipLaunchKernelGGL((adamw_chunked_kernel<TP, TSP>), grid, dim3(256), 0,   \
        stream, ptrs, slab_tensor, slab_idx, numels, T_, (float)lr,           \
        (float)beta1, (float)beta2, (float)eps, wd_factor, step_size,         \
        bc2_rsqrt)
    if (dt == torch::kBFloat16 && st_dt == torch::kFloat) {
        LAUNCH_ADAMW(__hip_bfloat16, float);
    } else if (dt == torch::kBFloat16 && st_dt == torch::kBFloat16) {
        LAUNCH_ADAMW(__hip_bfloat16, __hip_bfloat16);
    } else {
        TORCH_CHECK(dt == torch::kFloat && st_dt == torch::kFloat,
                    "adamw_fused_step: p bf16/f32, state p-dtype or f32");
        LAUNCH_ADAMW(float, float);
    }
#undef LAUNCH_ADAMW
    HIP_CHECK_LAUNCH();
}

int64_t adamw_slab_elems() { return SLAB; }
