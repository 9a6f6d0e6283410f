// Fused LayerNorm forward/backward for gfx950.
//
// Memory-bound: target is the HBM roofline (one read + one write of x per
// pass).  bf16 loads are vectorized 8-wide (16 B/lane); stats accumulate in
// fp32; mean/rstd saved fp32 for the backward.
//
// Replaces the reference's plain nn.LayerNorm (pipegoose
// nn/tensor_parallel/layer_norm.py:23-25) on the GPU path.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

// One block per row. H must be a multiple of 8*? no: tail handled scalar.
template <typename T, int BLOCK>
__global__ void layer_norm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ y, float* __restrict__ mean_out, float* __restrict__ rstd_out,
    int H, float eps) {
    __shared__ float smem[BLOCK / WAVE_SIZE];
    const int64_t row = blockIdx.x;
    const T* xr = x + row * (int64_t)H;
    T* yr = y + row * (int64_t)H;

    using V = typename vec8<T>::type;
    const int HV = H / 8;
    // cache this thread's packets across the two phases: H <= CACHE*8*BLOCK
    // rows never re-read x from HBM (one read + one write per element)
    constexpr int CACHE = 4;
    V reg[CACHE];
    const bool cached = HV <= CACHE * BLOCK;

    float sum = 0.f, sumsq = 0.f;
    for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
        V pkt = reinterpret_cast<const V*>(xr)[i];
        if (cached) reg[c] = pkt;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = to_float(pkt.v[j]);
            sum += f;
            sumsq += f * f;
        }
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        float f = to_float(xr[i]);
        sum += f;
        sumsq += f * f;
    }
    sum = block_reduce_sum(sum, smem);
    sumsq = block_reduce_sum(sumsq, smem);

    const float mean = sum / H;
    const float var = sumsq / H - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
        mean_out[row] = mean;
        rstd_out[row] = rstd;
    }

    for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
        V pkt = cached ? reg[c] : reinterpret_cast<const V*>(xr)[i];
        V wp = reinterpret_cast<const V*>(w)[i];
        V bp = reinterpret_cast<const V*>(b)[i];
        V out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float xhat = (to_float(pkt.v[j]) - mean) * rstd;
            out.v[j] = from_float<T>(xhat * to_float(wp.v[j]) + to_float(bp.v[j]));
        }
        reinterpret_cast<V*>(yr)[i] = out;
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        float xhat = (to_float(xr[i]) - mean) * rstd;
        yr[i] = from_float<T>(xhat * to_float(w[i]) + to_float(b[i]));
    }
}

// dx = rstd * (dyw - mean(dyw) - xhat * mean(dyw * xhat)), dyw = dy * w
template <typename T, int BLOCK>
__global__ void layer_norm_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    T* __restrict__ dx, int H) {
    __shared__ float smem[BLOCK / WAVE_SIZE];
    const int64_t row = blockIdx.x;
    const T* dyr = dy + row * (int64_t)H;
    const T* xr = x + row * (int64_t)H;
    T* dxr = dx + row * (int64_t)H;
    const float mean = mean_in[row];
    const float rstd = rstd_in[row];

    using V = typename vec8<T>::type;
    const int HV = H / 8;

    float s1 = 0.f, s2 = 0.f;  // sum(dyw), sum(dyw * xhat)
    for (int i = threadIdx.x; i < HV; i += BLOCK) {
        V dyp = reinterpret_cast<const V*>(dyr)[i];
        V xp = reinterpret_cast<const V*>(xr)[i];
        V wp = reinterpret_cast<const V*>(w)[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float dyw = to_float(dyp.v[j]) * to_float(wp.v[j]);
            float xhat = (to_float(xp.v[j]) - mean) * rstd;
            s1 += dyw;
            s2 += dyw * xhat;
        }
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        float dyw = to_float(dyr[i]) * to_float(w[i]);
        float xhat = (to_float(xr[i]) - mean) * rstd;
        s1 += dyw;
        s2 += dyw * xhat;
    }
    s1 = block_reduce_sum(s1, smem) / H;
    s2 = block_reduce_sum(s2, smem) / H;

    for (int i = threadIdx.x; i < HV; i += BLOCK) {
        V dyp = reinterpret_cast<const V*>(dyr)[i];
        V xp = reinterpret_cast<const V*>(xr)[i];
        V wp = reinterpret_cast<const V*>(w)[i];
        V out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float dyw = to_float(dyp.v[j]) * to_float(wp.v[j]);
            float xhat = (to_float(xp.v[j]) - mean) * rstd;
            out.v[j] = from_float<T>(rstd * (dyw - s1 - xhat * s2));
        }
        reinterpret_cast<V*>(dxr)[i] = out;
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        float dyw = to_float(dyr[i]) * to_float(w[i]);
        float xhat = (to_float(xr[i]) - mean) * rstd;
        dxr[i] = from_float<T>(rstd * (dyw - s1 - xhat * s2));
    }
}

// dw[j] = sum_i dy[i,j] * xhat[i,j]; db[j] = sum_i dy[i,j]
// Column-parallel, 8 columns per thread (vectorized 16-B bf16 loads — G13:
// scalar bf16 column reads were 8x off the HBM roofline).  Thread t owns
// columns [8*(blockIdx.x*BLOCK + t), +8); rows strided over gridDim.y.
template <typename T, int BLOCK>
__global__ void layer_norm_bwd_dwdb_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    float* __restrict__ dw, float* __restrict__ db, int64_t N, int H) {
    using V = typename vec8<T>::type;
    const int colv = blockIdx.x * BLOCK + threadIdx.x;  // packet column index
    if (colv >= H / 8) return;
    float sw[8] = {0.f}, sb[8] = {0.f};
    const int HV = H / 8;
    for (int64_t i = blockIdx.y; i < N; i += gridDim.y) {
        const float mean = mean_in[i];
        const float rstd = rstd_in[i];
        V dyp = reinterpret_cast<const V*>(dy + i * H)[colv];
        V xp = reinterpret_cast<const V*>(x + i * H)[colv];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float dyv = to_float(dyp.v[j]);
            float xhat = (to_float(xp.v[j]) - mean) * rstd;
            sw[j] += dyv * xhat;
            sb[j] += dyv;
        }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        // partials per y-block: fp32 atomics on shared columns serialize
        // (512 blocks -> ~130us, 4096 -> ~940us measured); a two-stage
        // partial+reduce keeps everything coalesced and contention-free
        const int col = colv * 8 + j;
        dw[(int64_t)blockIdx.y * H + col] = sw[j];
        db[(int64_t)blockIdx.y * H + col] = sb[j];
    }
}

// stage 2: column-sum the [GY, H] partials.  blockIdx.y splits the GY
// range (16 partials per block) so small H still spawns enough threads;
// each column gets GY/16 atomics — negligible contention.
__global__ void colsum_reduce_kernel(const float* __restrict__ part_w,
                                     const float* __restrict__ part_b,
                                     float* __restrict__ dw,
                                     float* __restrict__ db, int GY, int H) {
    const int col = blockIdx.x * blockDim.x + threadIdx.x;
    if (col >= H) return;
    const int g0 = blockIdx.y * 16;
    const int g1 = min(g0 + 16, GY);
    float aw = 0.f, ab = 0.f;
    for (int g = g0; g < g1; ++g) {
        aw += part_w[(int64_t)g * H + col];
        ab += part_b[(int64_t)g * H + col];
    }
    if (gridDim.y == 1) {
        dw[col] = aw;
        db[col] = ab;
    } else {
        atomicAdd(&dw[col], aw);
        atomicAdd(&db[col], ab);
    }
}

// scalar fallback when H % 8 != 0
template <typename T, int BLOCK>
__global__ void layer_norm_bwd_dwdb_scalar_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    float* __restrict__ dw, float* __restrict__ db, int64_t N, int H) {
    const int col = blockIdx.x * BLOCK + threadIdx.x;
    if (col >= H) return;
    float sw = 0.f, sb = 0.f;
    for (int64_t i = blockIdx.y; i < N; i += gridDim.y) {
        float dyv = to_float(dy[i * H + col]);
        float xhat = (to_float(x[i * H + col]) - mean_in[i]) * rstd_in[i];
        sw += dyv * xhat;
        sb += dyv;
    }
    if (gridDim.y == 1) {
        dw[col] = sw;
        db[col] = sb;
    } else {
        atomicAdd(&dw[col], sw);
        atomicAdd(&db[col], sb);
    }
}

}  // namespace

std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x, torch::Tensor w,
                                          torch::Tensor b, double eps) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous());
    const int64_t H = x.size(-1);
    const int64_t N = x.numel() / H;
    auto y = torch::empty_like(x);
    auto mean = torch::empty({N}, x.options().dtype(torch::kFloat));
    auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat));

    constexpr int BLOCK = 256;
    dim3 grid(N);
    auto stream = at::cuda::getCurrentCUDAStream();
    if (x.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((layer_norm_fwd_kernel<__hip_bfloat16, BLOCK>), grid, dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(b.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
            mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)H, (float)eps);
    } else if (x.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((layer_norm_fwd_kernel<float, BLOCK>), grid, dim3(BLOCK), 0, stream,
            x.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),
            y.data_ptr<float>(),
            mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)H, (float)eps);
    } else {
        TORCH_CHECK(false, "layer_norm_fwd: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return {y, mean, rstd};
}

std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor w, torch::Tensor mean,
                                          torch::Tensor rstd) {
    TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
    const int64_t H = x.size(-1);
    const int64_t N = x.numel() / H;
    auto dx = torch::empty_like(x);
    auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat));
    auto db = torch::zeros({H}, x.options().dtype(torch::kFloat));
    torch::Tensor part_w, part_b;

    constexpr int BLOCK = 256;
    auto stream = at::cuda::getCurrentCUDAStream();
    // dwdb: fill the chip (256 CUs, >=2 blocks/CU) with row-strided blocks;
    // fp32 atomics combine partials (512 adds/column max — low contention).
    const bool vec = (H % 8 == 0);
    const int64_t cols = vec ? H / 8 : H;
    int grid_y = (int)std::min<int64_t>(N, 256);
    dim3 grid_dw((cols + BLOCK - 1) / BLOCK, grid_y);
    if (vec) {
        part_w = torch::empty({grid_y, H}, x.options().dtype(torch::kFloat));
        part_b = torch::empty({grid_y, H}, x.options().dtype(torch::kFloat));
    }
    dim3 grid_red((H + BLOCK - 1) / BLOCK, (grid_y + 15) / 16);

    if (x.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((layer_norm_bwd_dx_kernel<__hip_bfloat16, BLOCK>), dim3(N), dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()), (int)H);
        if (vec) {
            hipLaunchKernelGGL((layer_norm_bwd_dwdb_kernel<__hip_bfloat16, BLOCK>), grid_dw, dim3(BLOCK), 0, stream,
                reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                mean.data_ptr<float>(), rstd.data_ptr<float>(),
                part_w.data_ptr<float>(), part_b.data_ptr<float>(), N, (int)H);
            hipLaunchKernelGGL((colsum_reduce_kernel), grid_red, dim3(BLOCK), 0, stream,
                part_w.data_ptr<float>(), part_b.data_ptr<float>(),
                dw.data_ptr<float>(), db.data_ptr<float>(), grid_y, (int)H);
        } else {
            hipLaunchKernelGGL((layer_norm_bwd_dwdb_scalar_kernel<__hip_bfloat16, BLOCK>), grid_dw, dim3(BLOCK), 0, stream,
                reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                mean.data_ptr<float>(), rstd.data_ptr<float>(),
                dw.data_ptr<float>(), db.data_ptr<float>(), N, (int)H);
        }
    } else if (x.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((layer_norm_bwd_dx_kernel<float, BLOCK>), dim3(N), dim3(BLOCK), 0, stream,
            dy.data_ptr<float>(), x.data_ptr<float>(), w.data_ptr<float>(),
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            dx.data_ptr<float>(), (int)H);
        if (vec) {
            hipLaunchKernelGGL((layer_norm_bwd_dwdb_kernel<float, BLOCK>), grid_dw, dim3(BLOCK), 0, stream,
                dy.data_ptr<float>(), x.data_ptr<float>(),
                mean.data_ptr<float>(), rstd.data_ptr<float>(),
                part_w.data_ptr<float>(), part_b.data_ptr<float>(), N, (int)H);
            hipLaunchKernelGGL((colsum_reduce_kernel), grid_red, dim3(BLOCK), 0, stream,
                part_w.data_ptr<float>(), part_b.data_ptr<float>(),
                dw.data_ptr<float>(), db.data_ptr<float>(), grid_y, (int)H);
        } else {
            hipLaunchKernelGGL((layer_norm_bwd_dwdb_scalar_kernel<float, BLOCK>), grid_dw, dim3(BLOCK), 0, stream,
                dy.data_ptr<float>(), x.data_ptr<float>(),
                mean.data_ptr<float>(), rstd.data_ptr<float>(),
                dw.data_ptr<float>(), db.data_ptr<float>(), N, (int)H);
        }
    } else {
        TORCH_CHECK(false, "layer_norm_bwd: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return {dx, dw.to(x.scalar_type()), db.to(x.scalar_type())};
}

// ------------------------------------------------------------------------
// Residual-fused forward: s = x + r; y = LN(s).  One HBM pass instead of a
// separate elementwise add kernel (the add showed up as ~2.6% of step time,
// profiles/bloom560m_1gpu_kernel_stats_r01.md); backward reuses
// layer_norm_bwd with x := s.
namespace {

template <typename T, int BLOCK>
__global__ void layer_norm_res_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ r,
    const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ s, T* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    int H, float eps) {
    __shared__ float smem[BLOCK / WAVE_SIZE];
    const int64_t row = blockIdx.x;
    const T* xr = x + row * (int64_t)H;
    const T* rr = r + row * (int64_t)H;
    T* sr = s + row * (int64_t)H;
    T* yr = y + row * (int64_t)H;

    using V = typename vec8<T>::type;
    const int HV = H / 8;
    constexpr int CACHE = 4;
    V reg[CACHE];
    const bool cached = HV <= CACHE * BLOCK;

    float sum = 0.f, sumsq = 0.f;
    for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
        V px = reinterpret_cast<const V*>(xr)[i];
        V pr = reinterpret_cast<const V*>(rr)[i];
        V ps;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = to_float(px.v[j]) + to_float(pr.v[j]);
            ps.v[j] = from_float<T>(f);
            // recompute from the rounded store so saved s EXACTLY matches
            // what backward will read (bit-stable for bf16)
            float fs = to_float(ps.v[j]);
            sum += fs;
            sumsq += fs * fs;
        }
        reinterpret_cast<V*>(sr)[i] = ps;
        if (cached) reg[c] = ps;
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        T sv = from_float<T>(to_float(xr[i]) + to_float(rr[i]));
        sr[i] = sv;
        float f = to_float(sv);
        sum += f;
        sumsq += f * f;
    }
    sum = block_reduce_sum(sum, smem);
    sumsq = block_reduce_sum(sumsq, smem);

    const float mean = sum / H;
    const float var = sumsq / H - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
        mean_out[row] = mean;
        rstd_out[row] = rstd;
    }

    for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
        V ps = cached ? reg[c] : reinterpret_cast<const V*>(sr)[i];
        V wp = reinterpret_cast<const V*>(w)[i];
        V bp = reinterpret_cast<const V*>(b)[i];
        V out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float xhat = (to_float(ps.v[j]) - mean) * rstd;
            out.v[j] = from_float<T>(xhat * to_float(wp.v[j]) + to_float(bp.v[j]));
        }
        reinterpret_cast<V*>(yr)[i] = out;
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        float xhat = (to_float(sr[i]) - mean) * rstd;
        yr[i] = from_float<T>(xhat * to_float(w[i]) + to_float(b[i]));
    }
}

}  // namespace

std::vector<torch::Tensor> layer_norm_res_fwd(torch::Tensor x, torch::Tensor r,
                                              torch::Tensor w, torch::Tensor b,
                                              double eps) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && r.is_contiguous());
    TORCH_CHECK(x.sizes() == r.sizes());
    const int64_t H = x.size(-1);
    const int64_t N = x.numel() / H;
    auto s = torch::empty_like(x);
    auto y = torch::empty_like(x);
    auto mean = torch::empty({N}, x.options().dtype(torch::kFloat));
    auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat));

    constexpr int BLOCK = 256;
    dim3 grid(N);
    auto stream = at::cuda::getCurrentCUDAStream();
    if (x.scalar_type() == torch::kBFloat16) {
        hipLaunchKernelGGL((layer_norm_res_fwd_kernel<__hip_bfloat16, BLOCK>), grid, dim3(BLOCK), 0, stream,
            reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(r.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
            reinterpret_cast<const __hip_bfloat16*>(b.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(s.data_ptr()),
            reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
            mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)H, (float)eps);
    } else if (x.scalar_type() == torch::kFloat) {
        hipLaunchKernelGGL((layer_norm_res_fwd_kernel<float, BLOCK>), grid, dim3(BLOCK), 0, stream,
            x.data_ptr<float>(), r.data_ptr<float>(),
            w.data_ptr<float>(), b.data_ptr<float>(),
            s.data_ptr<float>(), y.data_ptr<float>(),
            mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)H, (float)eps);
    } else {
        TORCH_CHECK(false, "layer_norm_res_fwd: unsupported dtype");
    }
    HIP_CHECK_LAUNCH();
    return {y, s, mean, rstd};
}
