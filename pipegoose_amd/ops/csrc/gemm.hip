// Hand-written CDNA4 MFMA GEMM for the TP-linear shapes (gfx950).
//
// C[M,N] = A[M,K] @ B[N,K]^T (+bias, optional gelu) in bf16 with fp32
// accumulation — the forward shape of Column/RowParallelLinear
// (nn/tensor_parallel/linear.py; reference semantics
// pipegoose/nn/tensor_parallel/linear.py:40-82).  Both operands are
// K-contiguous, which is the natural MFMA fragment orientation.
//
// Structure: the guide's "step-3" ladder rung — 128x128 tile, BK=64,
// 4 waves (2x2 of 64x64), double-buffered LDS staged by
// global_load_lds (16-B width, direct HBM->LDS, no VGPR round trip),
// XCD-aware bijective block remap for L2 affinity.  This is the
// match-or-fallback candidate against hipBLASLt:
// tools/gemm_bench.py produces the per-shape table; the dispatcher in
// ops/gemm.py only routes shapes where this kernel measured >= parity.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using frag_ab = __attribute__((ext_vector_type(8))) __bf16;
using frag_cd = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ frag_cd MFMA16(frag_ab a, frag_ab b, frag_cd c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int NTHREADS = 256;  // 4 waves

template <bool GELU>
__device__ __forceinline__ float epilogue(float x) {
    if (!GELU) return x;
    // tanh-approx GeLU (matches ops/csrc/bias_gelu.hip)
    const float c0 = 0.7978845608028654f, c1 = 0.044715f;
    const float u = c0 * (x + c1 * x * x * x);
    return 0.5f * x * (1.0f + tanhf(u));
}

template <bool BIAS, bool GELU>
__global__ __launch_bounds__(NTHREADS)
void gemm_bt_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                    const float* __restrict__ bias, bf16* __restrict__ C,
                    int M, int N, int K, int n_tiles_n) {
    // XCD-aware bijective remap (T1): give each XCD a contiguous chunk of
    // tile ids so neighbouring tiles (same A panel) share one L2.
    int wgid = blockIdx.x;
    {
        const int nwg = gridDim.x;
        const int q = nwg >> 3, r = nwg & 7;
        const int xcd = wgid & 7, idx = wgid >> 3;
        if (q > 0) {
            wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
        }
    }
    const int tm = wgid / n_tiles_n;
    const int tn = wgid % n_tiles_n;

    __shared__ __attribute__((aligned(16))) bf16 a_t[2][BM * BK];
    __shared__ __attribute__((aligned(16))) bf16 b_t[2][BN * BK];

    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lgrp = lane >> 4, lcol = lane & 15;
    const int wr = wave >> 1, wn = wave & 1;  // 2x2 waves of 64x64

    frag_cd acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = frag_cd{0.f, 0.f, 0.f, 0.f};

    // Each wave stages 4 KiB of each operand per K-tile: 4 glds of 1 KiB
    // (8 rows x 128 B).  chunk c covers rows [8c, 8c+8); lane l supplies
    // the global source for bytes [16l, 16l+16) of the chunk.
    const int64_t a_row0 = (int64_t)tm * BM;
    const int64_t b_row0 = (int64_t)tn * BN;
    auto stage = [&](int buf, int kt) {
        const int64_t k0 = (int64_t)kt * BK;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int c = wave * 4 + i;
            const int row = c * 8 + lane / 8;
            const int col = (lane % 8) * 8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)(
                    (const void*)(A + (a_row0 + row) * K + k0 + col)),
                (__attribute__((address_space(3))) unsigned int*)(
                    (void*)&a_t[buf][c * 512]),
                16, 0, 0);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)(
                    (const void*)(B + (b_row0 + row) * K + k0 + col)),
                (__attribute__((address_space(3))) unsigned int*)(
                    (void*)&b_t[buf][c * 512]),
                16, 0, 0);
        }
    };

    const int n_kt = K / BK;
    stage(0, 0);
    __syncthreads();  // drains the glds (compiler emits vmcnt(0) here)

    int buf = 0;
    for (int kt = 0; kt < n_kt; ++kt) {
        if (kt + 1 < n_kt) stage(buf ^ 1, kt + 1);
#pragma unroll
        for (int kk = 0; kk < BK / 32; ++kk) {
            frag_ab av[4];
#pragma unroll
            for (int mf = 0; mf < 4; ++mf) {
                av[mf] = *reinterpret_cast<const frag_ab*>(
                    &a_t[buf][(wr * 64 + mf * 16 + lcol) * BK + kk * 32 + 8 * lgrp]);
            }
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int nf = 0; nf < 4; ++nf) {
                frag_ab bv = *reinterpret_cast<const frag_ab*>(
                    &b_t[buf][(wn * 64 + nf * 16 + lcol) * BK + kk * 32 + 8 * lgrp]);
#pragma unroll
                for (int mf = 0; mf < 4; ++mf) {
                    acc[mf][nf] = MFMA16(av[mf], bv, acc[mf][nf]);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
        buf ^= 1;
    }

    // epilogue: C[row][col], row = 4*lgrp + r within each 16x16 fragment
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
        const int64_t col = b_row0 + wn * 64 + nf * 16 + lcol;
        float bv = 0.f;
        if (BIAS) bv = bias[col];
#pragma unroll
        for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int64_t row = a_row0 + wr * 64 + mf * 16 + 4 * lgrp + r;
                C[row * N + col] =
                    (bf16)epilogue<GELU>(acc[mf][nf][r] + bv);
            }
        }
    }
}

}  // namespace

torch::Tensor gemm_bt(torch::Tensor A, torch::Tensor B,
                      c10::optional<torch::Tensor> bias, bool gelu) {
    TORCH_CHECK(A.is_cuda() && A.dim() == 2 && B.dim() == 2);
    TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
                B.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
    const int M = A.size(0), K = A.size(1), N = B.size(0);
    TORCH_CHECK(B.size(1) == K, "gemm_bt: inner dims mismatch");
    TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
                "gemm_bt: M%128, N%128, K%64 must be 0");
    torch::Tensor bias_f;
    const bool has_bias = bias.has_value();
    if (has_bias) {
        bias_f = bias->to(torch::kFloat).contiguous();
        TORCH_CHECK(bias_f.numel() == N);
    }
    auto C = torch::empty({M, N}, A.options());
    const int n_tiles = (M / BM) * (N / BN);
    auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH(BIASV, GELUV)                                                  \
    hipLaunchKernelGGL((gemm_bt_kernel<BIASV, GELUV>), dim3(n_tiles),         \
        dim3(NTHREADS), 0, stream,                                            \
        reinterpret_cast<const bf16*>(A.data_ptr()),                          \
        reinterpret_cast<const bf16*>(B.data_ptr()),                          \
        has_bias ? bias_f.data_ptr<float>() : nullptr,                        \
        reinterpret_cast<bf16*>(C.data_ptr()), M, N, K, N / BN)
    if (has_bias && gelu) LAUNCH(true, true);
    else if (has_bias) LAUNCH(true, false);
    else if (gelu) LAUNCH(false, true);
    else LAUNCH(false, false);
#undef LAUNCH
    HIP_CHECK_LAUNCH();
    return C;
}
