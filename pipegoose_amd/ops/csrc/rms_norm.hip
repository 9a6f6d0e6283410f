// Fused RMSNorm forward/backward for gfx950 (north-star kernel list).
//
// y = x * rstd * w,  rstd = rsqrt(mean(x^2) + eps).  Same memory-bound
// structure as layer_norm.hip: one block per row, 8-wide bf16 vector loads,
// fp32 stats, rstd saved for backward.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

template <typename T, int BLOCK>
__global__ void rms_norm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w,
    T* __restrict__ y, float* __restrict__ rstd_out, int H, float eps) {
    __shared__ float smem[BLOCK / WAVE_SIZE];
    const int64_t row = blockIdx.x;
    const T* xr = x + row * (int64_t)H;
    T* yr = y + row * (int64_t)H;

    using V = typename vec8<T>::type;
    const int HV = H / 8;
    constexpr int CACHE = 4;
    V reg[CACHE];
    const bool cached = HV <= CACHE * BLOCK;

    float sumsq = 0.f;
    for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
        V pkt = reinterpret_cast<const V*>(xr)[i];
        if (cached) reg[c] = pkt;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = to_float(pkt.v[j]);
            sumsq += f * f;
        }
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        float f = to_float(xr[i]);
        sumsq += f * f;
    }
    sumsq = block_reduce_sum(sumsq, smem);
    const float rstd = rsqrtf(sumsq / H + eps);
    if (threadIdx.x == 0) rstd_out[row] = rstd;

    for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
        V pkt = cached ? reg[c] : reinterpret_cast<const V*>(xr)[i];
        V wp = reinterpret_cast<const V*>(w)[i];
        V out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            out.v[j] = from_float<T>(to_float(pkt.v[j]) * rstd * to_float(wp.v[j]));
        }
        reinterpret_cast<V*>(yr)[i] = out;
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        yr[i] = from_float<T>(to_float(xr[i]) * rstd * to_float(w[i]));
    }
}

// dx = rstd * (dyw - xhat * mean(dyw * xhat)),  dyw = dy*w, xhat = x*rstd
template <typename T, int BLOCK>
__global__ void rms_norm_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ rstd_in, T* __restrict__ dx, int H) {
    __shared__ float smem[BLOCK / WAVE_SIZE];
    const int64_t row = blockIdx.x;
    const T* dyr = dy + row * (int64_t)H;
    const T* xr = x + row * (int64_t)H;
    T* dxr = dx + row * (int64_t)H;
    const float rstd = rstd_in[row];

    float dot = 0.f;
    for (int i = threadIdx.x; i < H; i += BLOCK) {
        float dyw = to_float(dyr[i]) * to_float(w[i]);
        float xhat = to_float(xr[i]) * rstd;
        dot += dyw * xhat;
    }
    dot = block_reduce_sum(dot, smem) / H;

    for (int i = threadIdx.x; i < H; i += BLOCK) {
        float dyw = to_float(dyr[i]) * to_float(w[i]);
        float xhat = to_float(xr[i]) * rstd;
        dxr[i] = from_float<T>(rstd * (dyw - xhat * dot) * (1.0f));
    }
}

// dw[j] = sum_rows dy[r][j] * xhat[r][j] — per-y-block partials (atomics on
// shared columns serialize; two-stage partial+reduce is contention-free)
template <typename T, int BLOCK>
__global__ void rms_norm_bwd_dw_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ rstd_in, float* __restrict__ dw_part,
    int64_t N, int H) {
    const int col = blockIdx.x * BLOCK + threadIdx.x;
    if (col >= H) return;
    float acc = 0.f;
    for (int64_t r = blockIdx.y; r < N; r += gridDim.y) {
        acc += to_float(dy[r * H + col]) * to_float(x[r * H + col]) * rstd_in[r];
    }
    dw_part[(int64_t)blockIdx.y * H + col] = acc;
}

__global__ void rms_colsum_reduce_kernel(const float* __restrict__ part,
                                         float* __restrict__ dw,
                                         int GY, int H) {
    const int col = blockIdx.x * blockDim.x + threadIdx.x;
    if (col >= H) return;
    float acc = 0.f;
    for (int g = 0; g < GY; ++g) acc += part[(int64_t)g * H + col];
    dw[col] = acc;
}

}  // namespace

std::vector<torch::Tensor> rms_norm_fwd(torch::Tensor x, torch::Tensor w,
                                        double eps) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous());
    const int64_t H = x.size(-1);
    const int64_t N = x.numel() / H;
    auto y = torch::empty_like(x);
    auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat));

    constexpr int BLOCK = 256;
    auto stream = at::cuda::getCurrentCUDAStream();
    if (x.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((rms_norm_fwd_kernel<__hip_bfloat16, BLOCK>),
            dim3(N), dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
            rstd.data_ptr<float>(), (int)H, (float)eps);
    } else if (x.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((rms_norm_fwd_kernel<float, BLOCK>),
            dim3(N), dim3(BLOCK), 0, stream,
            x.data_ptr<float>(), w.data_ptr<float>(), y.data_ptr<float>(),
            rstd.data_ptr<float>(), (int)H, (float)eps);
    } else {
        TORCH_CHECK(false, "rms_norm_fwd: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return {y, rstd};
}

std::vector<torch::Tensor> rms_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w, torch::Tensor rstd) {
    TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
    const int64_t H = x.size(-1);
    const int64_t N = x.numel() / H;
    auto dx = torch::empty_like(x);
    auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat));

    constexpr int BLOCK = 256;
    auto stream = at::cuda::getCurrentCUDAStream();
    int grid_y = (int)std::min<int64_t>(N, 256);
    dim3 grid_dw((H + BLOCK - 1) / BLOCK, grid_y);
    auto part = torch::empty({grid_y, H}, x.options().dtype(torch::kFloat));
    dim3 grid_red((H + BLOCK - 1) / BLOCK);
    if (x.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((rms_norm_bwd_dx_kernel<__hip_bfloat16, BLOCK>),
            dim3(N), dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
            rstd.data_ptr<float>(),
            reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()), (int)H);
        hipLaunchKernelGGL((rms_norm_bwd_dw_kernel<__hip_bfloat16, BLOCK>),
            grid_dw, dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
            rstd.data_ptr<float>(), part.data_ptr<float>(), N, (int)H);
        hipLaunchKernelGGL((rms_colsum_reduce_kernel), grid_red, dim3(BLOCK),
            0, stream, part.data_ptr<float>(), dw.data_ptr<float>(),
            grid_y, (int)H);
    } else if (x.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((rms_norm_bwd_dx_kernel<float, BLOCK>),
            dim3(N), dim3(BLOCK), 0, stream,
            dy.data_ptr<float>(), x.data_ptr<float>(), w.data_ptr<float>(),
            rstd.data_ptr<float>(), dx.data_ptr<float>(), (int)H);
        hipLaunchKernelGGL((rms_norm_bwd_dw_kernel<float, BLOCK>),
            grid_dw, dim3(BLOCK), 0, stream,
            dy.data_ptr<float>(), x.data_ptr<float>(),
            rstd.data_ptr<float>(), part.data_ptr<float>(), N, (int)H);
        hipLaunchKernelGGL((rms_colsum_reduce_kernel), grid_red, dim3(BLOCK),
            0, stream, part.data_ptr<float>(), dw.data_ptr<float>(),
            grid_y, (int)H);
    } else {
        TORCH_CHECK(false, "rms_norm_bwd: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return {dx, dw.to(x.scalar_type())};
}
