// Fused rotary position embedding (RoPE) for gfx950 (north-star kernel list).
//
// Half-split (NeoX/LLaMA) convention on [B, H, S, D] tensors:
//   y[..., :D/2] = x1*cos - x2*sin ;  y[..., D/2:] = x2*cos + x1*sin
// with angle(s, d) = s * theta^(-2d/D).  cos/sin are computed IN-KERNEL from
// the base frequency — no precomputed [S, D] cos/sin tensors read from HBM.
// Backward is the inverse rotation (sign flip) — same kernel.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

template <typename T>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            int S, int D, float theta_base, float sign,
                            int64_t total_rows, int pos_offset) {
    // one wave per (b,h,s) row; lanes cover the D/2 rotation pairs
    const int64_t row = (int64_t)blockIdx.x * (blockDim.x / WAVE_SIZE)
                      + threadIdx.x / WAVE_SIZE;
    if (row >= total_rows) return;
    const int lane = threadIdx.x % WAVE_SIZE;
    const int s = (int)(row % S) + pos_offset;
    const T* xr = x + row * D;
    T* yr = y + row * D;
    const int half = D / 2;
    for (int d = lane; d < half; d += WAVE_SIZE) {
        const float freq = __powf(theta_base, -2.0f * d / (float)D);
        const float ang = s * freq;
        float c, sn;
        __sincosf(ang, &sn, &c);
        sn *= sign;
        const float x1 = to_float(xr[d]);
        const float x2 = to_float(xr[d + half]);
        yr[d] = from_float<T>(x1 * c - x2 * sn);
        yr[d + half] = from_float<T>(x2 * c + x1 * sn);
    }
}

}  // namespace

torch::Tensor rope_apply(torch::Tensor x, double theta_base, bool backward,
                         int64_t pos_offset) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
    const int S = x.size(2), D = x.size(3);
    TORCH_CHECK(D % 2 == 0);
    const int64_t rows = x.numel() / D;
    auto y = torch::empty_like(x);
    constexpr int BLOCK = 256;
    const int waves_per_block = BLOCK / WAVE_SIZE;
    dim3 grid((rows + waves_per_block - 1) / waves_per_block);
    auto stream = at::cuda::getCurrentCUDAStream();
    const float sign = backward ? -1.0f : 1.0f;
    if (x.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((rope_kernel<__hip_bfloat16>), grid, dim3(BLOCK), 0,
            stream,
            reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
            S, D, (float)theta_base, sign, rows, (int)pos_offset);
    } else if (x.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((rope_kernel<float>), grid, dim3(BLOCK), 0, stream,
            x.data_ptr<float>(), y.data_ptr<float>(),
            S, D, (float)theta_base, sign, rows, (int)pos_offset);
    } else {
        TORCH_CHECK(false, "rope_apply: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return y;
}
