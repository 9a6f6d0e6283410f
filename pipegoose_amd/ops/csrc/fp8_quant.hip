// fp8 dynamic-quantization kernels, gfx950 (OCP e4m3fn / e5m2).
//
// The eager quantize (abs().amax(); t.float()/scale; clamp; .to(fp8)) costs
// 4-5 full fp32 HBM passes per tensor and made the fp8 linear path LOSE to
// bf16 despite the 2x GEMM rate (tools/fp8_linear_bench.py, first cut).
// Here: one bf16 read for the block-partial amax, a tiny finalize, and one
// read+write pass for the cast (hardware v_cvt float->fp8 via the
// __hip_fp8_* types — gfx950 is OCP, NOT the MI300 fnuz encoding), plus an
// LDS-tiled byte transpose replacing the strided .t().contiguous() copies
// the backward GEMM layouts need.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <hip/hip_fp8.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;

constexpr float E4M3_MAX = 448.0f;
constexpr float E5M2_MAX = 57344.0f;

struct bf16x8 {
    bf16 v[8];
};

__global__ void amax_partial_kernel(const bf16* __restrict__ x,
                                    float* __restrict__ partials, int64_t n) {
    __shared__ float smem[8];
    float m = 0.f;
    const int64_t nv = n / 8;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
         i += (int64_t)gridDim.x * blockDim.x) {
        bf16x8 xv = reinterpret_cast<const bf16x8*>(x)[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) m = fmaxf(m, fabsf(to_float(xv.v[j])));
    }
    // tail (n % 8) handled by block 0, thread 0..7
    if (blockIdx.x == 0 && threadIdx.x < n - nv * 8)
        m = fmaxf(m, fabsf(to_float(x[nv * 8 + threadIdx.x])));
    m = block_reduce_max(m, smem);
    if (threadIdx.x == 0) partials[blockIdx.x] = m;
}

// partials -> scale[0] = dequant (amax/max_val), scale[1] = quant multiplier
__global__ void finalize_scale_kernel(const float* __restrict__ partials,
                                      int np, float max_val,
                                      float* __restrict__ scale) {
    __shared__ float smem[8];
    float m = 0.f;
    for (int i = threadIdx.x; i < np; i += blockDim.x)
        m = fmaxf(m, partials[i]);
    m = block_reduce_max(m, smem);
    if (threadIdx.x == 0) {
        m = fmaxf(m, 1e-12f);
        scale[0] = m / max_val;
        scale[1] = max_val / m;
    }
}

template <bool E5M2>
__global__ void cast_kernel(const bf16* __restrict__ x,
                            uint8_t* __restrict__ q,
                            const float* __restrict__ scale, int64_t n) {
    const float inv = scale[1];
    const float maxv = E5M2 ? E5M2_MAX : E4M3_MAX;
    const int64_t nv = n / 8;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
         i += (int64_t)gridDim.x * blockDim.x) {
        bf16x8 xv = reinterpret_cast<const bf16x8*>(x)[i];
        union { uint8_t b[8]; uint2 u; } out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = fminf(fmaxf(to_float(xv.v[j]) * inv, -maxv), maxv);
            if (E5M2) {
                __hip_fp8_e5m2 c(f);
                out.b[j] = c.__x;
            } else {
                __hip_fp8_e4m3 c(f);
                out.b[j] = c.__x;
            }
        }
        reinterpret_cast<uint2*>(q)[i] = out.u;
    }
    if (blockIdx.x == 0 && threadIdx.x < n - nv * 8) {
        const int64_t i = nv * 8 + threadIdx.x;
        float f = fminf(fmaxf(to_float(x[i]) * inv, -maxv), maxv);
        if (E5M2) {
            __hip_fp8_e5m2 c(f);
            q[i] = c.__x;
        } else {
            __hip_fp8_e4m3 c(f);
            q[i] = c.__x;
        }
    }
}

// Fused cast + dual-layout write: one bf16 read produces BOTH the
// row-major and the transposed fp8 image (64x64 LDS tile for the
// transposed side).  Eliminates the separate byte-transpose pass the
// backward GEMM layouts otherwise need (measured 5.8% of the fp8-7b1
// step).  Requires C % 4 == 0 (uchar4 row-major stores).
template <bool E5M2>
__global__ void cast_dual_kernel(const bf16* __restrict__ in,
                                 uint8_t* __restrict__ q,
                                 uint8_t* __restrict__ qt,
                                 const float* __restrict__ scale,
                                 int R, int C) {
    __shared__ uint8_t tile[64][65];
    const float inv = scale[1];
    const float maxv = E5M2 ? E5M2_MAX : E4M3_MAX;
    const int tr = blockIdx.y * 64;
    const int tc = blockIdx.x * 64;
    const int t = threadIdx.x;            // 256 threads
    const int lr = t / 16, lc4 = (t % 16) * 4;
#pragma unroll
    for (int s = 0; s < 4; ++s) {
        const int r = tr + lr + s * 16;
        const int c = tc + lc4;
        if (r < R && c < C) {
            union { uint8_t b[4]; uchar4 u; } o;
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                float f = to_float(in[(int64_t)r * C + c + j]);
                f = fminf(fmaxf(f * inv, -maxv), maxv);
                if (E5M2) {
                    __hip_fp8_e5m2 cv(f);
                    o.b[j] = cv.__x;
                } else {
                    __hip_fp8_e4m3 cv(f);
                    o.b[j] = cv.__x;
                }
                tile[lr + s * 16][lc4 + j] = o.b[j];
            }
            *reinterpret_cast<uchar4*>(q + (int64_t)r * C + c) = o.u;
        }
    }
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {
        const int c = tc + lr + s * 16;   // output row = input column
        if (c < C) {
            const int r = tr + lc4;
            if (r + 3 < R) {
                uchar4 v;
                v.x = tile[lc4 + 0][lr + s * 16];
                v.y = tile[lc4 + 1][lr + s * 16];
                v.z = tile[lc4 + 2][lr + s * 16];
                v.w = tile[lc4 + 3][lr + s * 16];
                *reinterpret_cast<uchar4*>(qt + (int64_t)c * R + r) = v;
            } else {
                for (int j = 0; j < 4 && r + j < R; ++j)
                    qt[(int64_t)c * R + r + j] = tile[lc4 + j][lr + s * 16];
            }
        }
    }
}

// [R, C] bytes -> [C, R]: 64x64 LDS tiles, uchar4 global loads AND stores
// (both sides coalesced; the torch fallback is a strided byte copy).
__global__ void transpose_u8_kernel(const uint8_t* __restrict__ in,
                                    uint8_t* __restrict__ out,
                                    int R, int C) {
    __shared__ uint8_t tile[64][65];
    const int tr = blockIdx.y * 64;
    const int tc = blockIdx.x * 64;
    const int t = threadIdx.x;            // 256 threads
    const int lr = t / 16, lc4 = (t % 16) * 4;
#pragma unroll
    for (int s = 0; s < 4; ++s) {
        const int r = tr + lr + s * 16;
        if (r < R) {
            const int c = tc + lc4;
            if (c + 3 < C) {
                const uchar4 v = *reinterpret_cast<const uchar4*>(
                    in + (int64_t)r * C + c);
                tile[lr + s * 16][lc4 + 0] = v.x;
                tile[lr + s * 16][lc4 + 1] = v.y;
                tile[lr + s * 16][lc4 + 2] = v.z;
                tile[lr + s * 16][lc4 + 3] = v.w;
            } else {
                for (int j = 0; j < 4 && c + j < C; ++j)
                    tile[lr + s * 16][lc4 + j] = in[(int64_t)r * C + c + j];
            }
        }
    }
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {
        const int c = tc + lr + s * 16;   // output row = input column
        if (c < C) {
            const int r = tr + lc4;
            if (r + 3 < R) {
                uchar4 v;
                v.x = tile[lc4 + 0][lr + s * 16];
                v.y = tile[lc4 + 1][lr + s * 16];
                v.z = tile[lc4 + 2][lr + s * 16];
                v.w = tile[lc4 + 3][lr + s * 16];
                *reinterpret_cast<uchar4*>(out + (int64_t)c * R + r) = v;
            } else {
                for (int j = 0; j < 4 && r + j < R; ++j)
                    out[(int64_t)c * R + r + j] = tile[lc4 + j][lr + s * 16];
            }
        }
    }
}

}  // namespace

// Returns {q (fp8 tensor, same shape), scale[2] = {dequant, quant-mult}}.
std::vector<torch::Tensor> fp8_quant(torch::Tensor x, bool e5m2) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
                x.is_contiguous(), "fp8_quant: contiguous bf16 CUDA only");
    auto q = torch::empty_like(
        x, x.options().dtype(e5m2 ? torch::kFloat8_e5m2
                                  : torch::kFloat8_e4m3fn));
    auto scale = torch::empty({2}, x.options().dtype(torch::kFloat));
    const int64_t n = x.numel();
    const int threads = 256;
    const int blocks = (int)std::min<int64_t>((n / 8 + threads - 1) / threads,
                                              2048);
    auto partials = torch::empty({std::max(blocks, 1)},
                                 x.options().dtype(torch::kFloat));
    auto stream = at::cuda::getCurrentCUDAStream();
    const float maxv = e5m2 ? E5M2_MAX : E4M3_MAX;
    const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
    hipLaunchKernelGGL(amax_partial_kernel, dim3(std::max(blocks, 1)),
                       dim3(threads), 0, stream, xp,
                       partials.data_ptr<float>(), n);
    hipLaunchKernelGGL(finalize_scale_kernel, dim3(1), dim3(threads), 0,
                       stream, partials.data_ptr<float>(),
                       std::max(blocks, 1), maxv, scale.data_ptr<float>());
    uint8_t* qp = reinterpret_cast<uint8_t*>(q.data_ptr());
    if (e5m2)
        hipLaunchKernelGGL((cast_kernel<true>), dim3(std::max(blocks, 1)),
                           dim3(threads), 0, stream, xp, qp,
                           scale.data_ptr<float>(), n);
    else
        hipLaunchKernelGGL((cast_kernel<false>), dim3(std::max(blocks, 1)),
                           dim3(threads), 0, stream, xp, qp,
                           scale.data_ptr<float>(), n);
    return {q, scale};
}

// Returns {q [R,C], qT [C,R], scale[2]} from one bf16 read.
std::vector<torch::Tensor> fp8_quant_dual(torch::Tensor x, bool e5m2) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
                x.dim() == 2 && x.is_contiguous() && x.size(1) % 4 == 0,
                "fp8_quant_dual: contiguous 2D bf16, cols % 4 == 0");
    const int R = (int)x.size(0), C = (int)x.size(1);
    auto opts8 = x.options().dtype(e5m2 ? torch::kFloat8_e5m2
                                        : torch::kFloat8_e4m3fn);
    auto q = torch::empty({R, C}, opts8);
    auto qt = torch::empty({C, R}, opts8);
    auto scale = torch::empty({2}, x.options().dtype(torch::kFloat));
    const int64_t n = x.numel();
    const int threads = 256;
    const int blocks = (int)std::min<int64_t>((n / 8 + threads - 1) / threads,
                                              2048);
    auto partials = torch::empty({std::max(blocks, 1)},
                                 x.options().dtype(torch::kFloat));
    auto stream = at::cuda::getCurrentCUDAStream();
    const float maxv = e5m2 ? E5M2_MAX : E4M3_MAX;
    const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
    hipLaunchKernelGGL(amax_partial_kernel, dim3(std::max(blocks, 1)),
                       dim3(threads), 0, stream, xp,
                       partials.data_ptr<float>(), n);
    hipLaunchKernelGGL(finalize_scale_kernel, dim3(1), dim3(threads), 0,
                       stream, partials.data_ptr<float>(),
                       std::max(blocks, 1), maxv, scale.data_ptr<float>());
    dim3 grid((C + 63) / 64, (R + 63) / 64);
    uint8_t* qp = reinterpret_cast<uint8_t*>(q.data_ptr());
    uint8_t* qtp = reinterpret_cast<uint8_t*>(qt.data_ptr());
    if (e5m2)
        hipLaunchKernelGGL((cast_dual_kernel<true>), grid, dim3(256), 0,
                           stream, xp, qp, qtp, scale.data_ptr<float>(), R, C);
    else
        hipLaunchKernelGGL((cast_dual_kernel<false>), grid, dim3(256), 0,
                           stream, xp, qp, qtp, scale.data_ptr<float>(), R, C);
    return {q, qt, scale};
}

torch::Tensor fp8_transpose(torch::Tensor q) {
    TORCH_CHECK(q.is_cuda() && q.dim() == 2 && q.is_contiguous() &&
                q.element_size() == 1, "fp8_transpose: contiguous 2D bytes");
    const int R = (int)q.size(0), C = (int)q.size(1);
    auto out = torch::empty({C, R}, q.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(transpose_u8_kernel,
                       dim3((C + 63) / 64, (R + 63) / 64), dim3(256), 0,
                       stream,
                       reinterpret_cast<const uint8_t*>(q.data_ptr()),
                       reinterpret_cast<uint8_t*>(out.data_ptr()), R, C);
    return out;
}
