// Fused SwiGLU elementwise: y = silu(gate) * up, gfx950.
//
// One HBM pass instead of eager's three (sigmoid-mul, mul, and the silu
// intermediate); backward fuses both input grads in one pass.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

__device__ __forceinline__ float silu(float x) {
    return x / (1.0f + __expf(-x));
}

template <typename T>
__global__ void silu_mul_fwd_kernel(const T* __restrict__ g,
                                    const T* __restrict__ u,
                                    T* __restrict__ y, int64_t n) {
    using V = typename vec8<T>::type;
    const int64_t nv = n / 8;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
         i += (int64_t)gridDim.x * blockDim.x) {
        V gv = reinterpret_cast<const V*>(g)[i];
        V uv = reinterpret_cast<const V*>(u)[i];
        V out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            out.v[j] = from_float<T>(silu(to_float(gv.v[j])) * to_float(uv.v[j]));
        }
        reinterpret_cast<V*>(y)[i] = out;
    }
}

// dgate = dy * up * dsilu(gate);  dup = dy * silu(gate)
template <typename T>
__global__ void silu_mul_bwd_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ g,
                                    const T* __restrict__ u,
                                    T* __restrict__ dg, T* __restrict__ du,
                                    int64_t n) {
    using V = typename vec8<T>::type;
    const int64_t nv = n / 8;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
         i += (int64_t)gridDim.x * blockDim.x) {
        V dyv = reinterpret_cast<const V*>(dy)[i];
        V gv = reinterpret_cast<const V*>(g)[i];
        V uv = reinterpret_cast<const V*>(u)[i];
        V dgo, duo;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float x = to_float(gv.v[j]);
            float s = 1.0f / (1.0f + __expf(-x));
            float sx = x * s;                       // silu(x)
            float dsilu = s * (1.0f + x * (1.0f - s));
            float d = to_float(dyv.v[j]);
            dgo.v[j] = from_float<T>(d * to_float(uv.v[j]) * dsilu);
            duo.v[j] = from_float<T>(d * sx);
        }
        reinterpret_cast<V*>(dg)[i] = dgo;
        reinterpret_cast<V*>(du)[i] = duo;
    }
}

}  // namespace

torch::Tensor silu_mul_fwd(torch::Tensor gate, torch::Tensor up) {
    TORCH_CHECK(gate.is_cuda() && gate.is_contiguous() && up.is_contiguous());
    TORCH_CHECK(gate.numel() == up.numel() && gate.numel() % 8 == 0);
    auto y = torch::empty_like(gate);
    const int64_t n = gate.numel();
    constexpr int BLOCK = 256;
    int grid = (int)std::min<int64_t>((n / 8 + BLOCK - 1) / BLOCK, 2048);
    auto stream = at::cuda::getCurrentCUDAStream();
    if (gate.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((silu_mul_fwd_kernel<__hip_bfloat16>), dim3(grid),
            dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(gate.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(up.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(y.data_ptr()), n);
    } else if (gate.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((silu_mul_fwd_kernel<float>), dim3(grid),
            dim3(BLOCK), 0, stream,
            gate.data_ptr<float>(), up.data_ptr<float>(), y.data_ptr<float>(), n);
    } else {
        TORCH_CHECK(false, "silu_mul_fwd: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return y;
}

std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dy, torch::Tensor gate,
                                        torch::Tensor up) {
    TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
    auto dg = torch::empty_like(gate);
    auto du = torch::empty_like(up);
    const int64_t n = gate.numel();
    constexpr int BLOCK = 256;
    int grid = (int)std::min<int64_t>((n / 8 + BLOCK - 1) / BLOCK, 2048);
    auto stream = at::cuda::getCurrentCUDAStream();
    if (gate.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((silu_mul_bwd_kernel<__hip_bfloat16>), dim3(grid),
            dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(gate.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(up.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(dg.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(du.data_ptr()), n);
    } else if (gate.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((silu_mul_bwd_kernel<float>), dim3(grid),
            dim3(BLOCK), 0, stream,
            dy.data_ptr<float>(), gate.data_ptr<float>(), up.data_ptr<float>(),
            dg.data_ptr<float>(), du.data_ptr<float>(), n);
    } else {
        TORCH_CHECK(false, "silu_mul_bwd: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return {dg, du};
}
