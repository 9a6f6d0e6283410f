// Flash attention (causal + ALiBi) for gfx950 — hand-written CDNA4 MFMA.
//
// Replaces the sdpa-with-bias fallback: the ALiBi bias slope*(j-i) is
// computed IN-KERNEL from per-head slopes, so no [H,S,S] bias tensor is
// materialized or re-read from HBM (that tensor dominated the sdpa path's
// memory traffic).  Online-softmax (flash2) with fp32 running stats.
//
// Tiling: one workgroup = 4 waves = 256 threads handles a 64-row Q block;
// each wave owns 16 q rows.  K/V tiles (64 rows) stage through LDS; V is
// stored transposed so the P·V MFMA's B-fragment reads are contiguous 16B.
// MFMA: v_mfma_f32_16x16x32_bf16 (A/B: 8 bf16/lane; C/D: 4 fp32/lane,
// row=(lane>>4)*4+reg, col=lane&15 — guide §3 fragment layout).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using frag_ab = __attribute__((ext_vector_type(8))) __bf16;   // 8 bf16 = 4 VGPR
using frag_cd = __attribute__((ext_vector_type(4))) float;    // 4 fp32

__device__ __forceinline__ frag_cd MFMA_16x16x32(frag_ab a, frag_ab b, frag_cd c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

constexpr int BLOCK_M = 64;   // q rows per workgroup
constexpr int BLOCK_N = 64;   // kv rows per LDS tile
constexpr int NWAVES = 4;     // BLOCK_M / 16
constexpr int PAD = 8;        // bf16 elements (16 B) of row padding in LDS
constexpr float NEG_INF = -1e30f;

// A-fragment (M=16 side): row = lane&15, k = 8*(lane>>4) + i
// B-fragment (N=16 side): col = lane&15, k = 8*(lane>>4) + i
// C/D:                    col = lane&15, row = 4*(lane>>4) + reg

template <int D, int MT, int WAVES = NWAVES>
__global__ __launch_bounds__(WAVES * WAVE_SIZE)
void attn_fwd_kernel(const bf16* __restrict__ q, const bf16* __restrict__ k,
                     const bf16* __restrict__ v,
                     const float* __restrict__ slopes, float scale,
                     bf16* __restrict__ o, float* __restrict__ lse,
                     int B, int H, int S, int kv_off, int kv_group,
                     int64_t qb, int64_t qh, int64_t qs,
                     int64_t kb, int64_t kh, int64_t ks,
                     int64_t vb, int64_t vh, int64_t vs,
                     int64_t ob, int64_t oh, int64_t os) {
    // Tensors are [B, H, S, D] LOGICAL with arbitrary strides (d contiguous):
    // the model's qkv views come straight from the fused projection with no
    // .contiguous() copies (those copies were ~6% of the training step).
    // MT = q row-tiles per wave (16 rows each).  MT=2 doubles the MFMA work
    // per LDS B-fragment read — the K/V tiles are read once per wave and
    // used for 32 q rows.
    constexpr int DCH = D / 32;    // K-chunks per QK^T mfma row
    constexpr int DSUB = D / 16;   // output d-subtiles
    constexpr int KSTRIDE = D + PAD;
    constexpr int VSTRIDE = BLOCK_N + PAD;
    constexpr int ROWS_PER_WG = WAVES * MT * 16;
    constexpr int NT = WAVES * WAVE_SIZE;

    __shared__ bf16 k_lds[BLOCK_N * KSTRIDE];
    __shared__ bf16 vt_lds[D * VSTRIDE];
    __shared__ bf16 p_lds[WAVES][MT * 16 * VSTRIDE];

    const int qblock = blockIdx.x;
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lgrp = lane >> 4;       // 0..3
    const int lcol = lane & 15;       // 0..15

    const int64_t bh_off = ((int64_t)b * H + h) * S;  // lse layout
    const int hk = h / kv_group;  // GQA: q heads share kv heads
    const bf16* qp = q + b * qb + h * qh;
    const bf16* kp = k + b * kb + hk * kh;
    const bf16* vp = v + b * vb + hk * vh;
    bf16* op = o + b * ob + h * oh;

    // wave owns rows [qrow0, qrow0 + MT*16)
    const int qrow0 = qblock * ROWS_PER_WG + wave * (MT * 16);
    const float slope = slopes[h];

    frag_ab aQ[MT][DCH];
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
        const bf16* qrow = qp + (int64_t)(qrow0 + mt * 16 + lcol) * qs;
#pragma unroll
        for (int c = 0; c < DCH; ++c) {
            aQ[mt][c] = *reinterpret_cast<const frag_ab*>(qrow + c * 32 + 8 * lgrp);
        }
    }

    frag_cd accO[MT][DSUB];
    float m_run[MT][4], l_run[MT][4];
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
        for (int s = 0; s < DSUB; ++s) accO[mt][s] = frag_cd{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int r = 0; r < 4; ++r) { m_run[mt][r] = NEG_INF; l_run[mt][r] = 0.f; }
    }

    // kv positions are globally shifted by kv_off (ring attention blocks):
    // kv_off <= -S means every kv position is visible (no causal cut)
    int n_kv_blocks = (kv_off <= -S) ? (S / BLOCK_N)
        : (qblock * ROWS_PER_WG + ROWS_PER_WG - kv_off + BLOCK_N - 1) / BLOCK_N;
    n_kv_blocks = min(max(n_kv_blocks, 0), S / BLOCK_N);
    for (int nb = 0; nb < n_kv_blocks; ++nb) {
        const int kvrow0 = nb * BLOCK_N;
        __syncthreads();
        {   // cooperative K tile load (row-major, 16B packets)
            constexpr int PACKETS = BLOCK_N * D / 8;
            for (int p = tid; p < PACKETS; p += NT) {
                const int row = p / (D / 8);
                const int col = (p % (D / 8)) * 8;
                *reinterpret_cast<frag_ab*>(&k_lds[row * KSTRIDE + col]) =
                    *reinterpret_cast<const frag_ab*>(
                        kp + (int64_t)(kvrow0 + row) * ks + col);
            }
        }
        {   // V tile load, transposed; two rows at once -> 4 B column stores
            constexpr int PACKETS = (BLOCK_N / 2) * (D / 8);
            for (int p = tid; p < PACKETS; p += NT) {
                const int row = (p / (D / 8)) * 2;
                const int col = (p % (D / 8)) * 8;
                frag_ab p0 = *reinterpret_cast<const frag_ab*>(
                    vp + (int64_t)(kvrow0 + row) * vs + col);
                frag_ab p1 = *reinterpret_cast<const frag_ab*>(
                    vp + (int64_t)(kvrow0 + row + 1) * vs + col);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    union { __bf16 h[2]; uint32_t u; } pair;
                    pair.h[0] = p0[j];
                    pair.h[1] = p1[j];
                    *reinterpret_cast<uint32_t*>(&vt_lds[(col + j) * VSTRIDE + row]) = pair.u;
                }
            }
        }
        __syncthreads();

#pragma unroll
        for (int mt = 0; mt < MT; ++mt) {
            // S = Q K^T for this row-tile (16 x 64)
            float s_tile[4][4];  // [nsub][reg]
#pragma unroll
            for (int ns = 0; ns < 4; ++ns) {
                frag_cd acc = frag_cd{0.f, 0.f, 0.f, 0.f};
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int c = 0; c < DCH; ++c) {
                    frag_ab bK = *reinterpret_cast<const frag_ab*>(
                        &k_lds[(ns * 16 + lcol) * KSTRIDE + c * 32 + 8 * lgrp]);
                    acc = MFMA_16x16x32(aQ[mt][c], bK, acc);
                }
                __builtin_amdgcn_s_setprio(0);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int iq = qrow0 + mt * 16 + 4 * lgrp + r;
                    const int jk = kvrow0 + ns * 16 + lcol + kv_off;
                    float sv = acc[r] * scale + slope * (float)(jk - iq);
                    s_tile[ns][r] = (jk <= iq) ? sv : NEG_INF;
                }
            }

            // online softmax update
            float m_new[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float mx = fmaxf(fmaxf(s_tile[0][r], s_tile[1][r]),
                                 fmaxf(s_tile[2][r], s_tile[3][r]));
#pragma unroll
                for (int off = 1; off < 16; off <<= 1) {
                    mx = fmaxf(mx, __shfl_xor(mx, off, WAVE_SIZE));
                }
                m_new[r] = fmaxf(m_run[mt][r], mx);
            }
            float rescale[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                rescale[r] = __expf(m_run[mt][r] - m_new[r]);
                m_run[mt][r] = m_new[r];
                l_run[mt][r] *= rescale[r];
            }
#pragma unroll
            for (int s = 0; s < DSUB; ++s) {
#pragma unroll
                for (int r = 0; r < 4; ++r) accO[mt][s][r] *= rescale[r];
            }

            // P = exp(S - m); stage bf16 for the PV mfma; accumulate l
#pragma unroll
            for (int ns = 0; ns < 4; ++ns) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    float pv = __expf(s_tile[ns][r] - m_run[mt][r]);
                    s_tile[ns][r] = pv;
                    p_lds[wave][(mt * 16 + 4 * lgrp + r) * VSTRIDE + ns * 16 + lcol] =
                        (bf16)pv;
                }
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float ls = s_tile[0][r] + s_tile[1][r] + s_tile[2][r] + s_tile[3][r];
#pragma unroll
                for (int off = 1; off < 16; off <<= 1) {
                    ls += __shfl_xor(ls, off, WAVE_SIZE);
                }
                l_run[mt][r] += ls;
            }
        }
        __builtin_amdgcn_s_waitcnt(0);  // own-wave P writes visible

        // O += P V  (K = 64 in 2 chunks of 32)
#pragma unroll
        for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
            for (int kc = 0; kc < 2; ++kc) {
                frag_ab aP = *reinterpret_cast<const frag_ab*>(
                    &p_lds[wave][(mt * 16 + lcol) * VSTRIDE + kc * 32 + 8 * lgrp]);
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int ds = 0; ds < DSUB; ++ds) {
                    frag_ab bV = *reinterpret_cast<const frag_ab*>(
                        &vt_lds[(ds * 16 + lcol) * VSTRIDE + kc * 32 + 8 * lgrp]);
                    accO[mt][ds] = MFMA_16x16x32(aP, bV, accO[mt][ds]);
                }
                __builtin_amdgcn_s_setprio(0);
            }
        }
    }

    // epilogue
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int iq = qrow0 + mt * 16 + 4 * lgrp + r;
            const float inv_l = 1.0f / l_run[mt][r];
            bf16* orow = op + (int64_t)iq * os;
#pragma unroll
            for (int ds = 0; ds < DSUB; ++ds) {
                orow[ds * 16 + lcol] = (bf16)(accO[mt][ds][r] * inv_l);
            }
            if (lcol == 0) {
                lse[bh_off + iq] = m_run[mt][r] + __logf(l_run[mt][r]);
            }
        }
    }
}

// ------------------------------------------------------------------ probe
// Verifies the assumed A/B fragment layouts: C[16][16] = A[16][32]·B[32][16]
// with A,B read from global memory exactly the way the attention kernel
// reads its fragments.  One wave.
__global__ void mfma_probe_kernel(const bf16* __restrict__ A,
                                  const bf16* __restrict__ Bt,
                                  float* __restrict__ C) {
    const int lane = threadIdx.x;
    const int lgrp = lane >> 4, lcol = lane & 15;
    // A[m][k]: m = lcol, k = 8*lgrp + i  (row-major A: 16x32)
    frag_ab a = *reinterpret_cast<const frag_ab*>(A + lcol * 32 + 8 * lgrp);
    // B[k][n] with Bt stored as [n][k] (row-major 16x32): n = lcol, k = 8*lgrp+i
    frag_ab b = *reinterpret_cast<const frag_ab*>(Bt + lcol * 32 + 8 * lgrp);
    frag_cd c = MFMA_16x16x32(a, b, frag_cd{0.f, 0.f, 0.f, 0.f});
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        C[(4 * lgrp + r) * 16 + lcol] = c[r];
    }
}

}  // namespace

// ======================================================================
// Forward v2 — the round-2 speed-of-light ladder (docs/KERNEL_PLAN_R2.md):
//   * 8 waves x 32 q rows, v_mfma_f32_32x32x16_bf16 (guide §B 8-warp ladder)
//   * swapped QK^T (St = K·Q^T): the P row for q-row lane&31 is lane-local
//     (32 f32 regs), so softmax is in-register — no p_lds bounce, no
//     __shfl tree (T12: cross-half combine = one permlane32_swap)
//   * swapped PV too (O^T = V^T·P^T): the O accumulator is ALSO per-lane
//     q-row lane&31, so the online-softmax rescale is a scalar broadcast
//     multiply — no cross-lane redistribution of alpha
//   * K and V^T LDS images on 256-B rows with the T2 XOR swizzle
//     (byte ^= (row&15)<<4): the b128 fragment reads are conflict-free
//     (v1's padded rows measured 33-63% LDS bank-conflict cycles)
//   * double-buffered tiles with the T14 split: global loads for tile n+1
//     issue before tile n's QK^T, the LDS write lands after softmax, so HBM
//     latency hides under MFMA; one barrier per tile
__device__ __forceinline__ float u2f(unsigned u) { return __uint_as_float(u); }

namespace {

using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;
using lds_bf16x4_p = __attribute__((address_space(3))) bf16x4*;

__device__ __forceinline__ f32x16 MFMA_32x32x16(frag_ab a, frag_ab b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

// Element index into a swizzled [row][128] bf16 image (256-B rows): the
// byte offset within the row is XORed with (row&15)<<4, spreading a b128
// lane group over all 16 slots of the 256-B bank row (T2 / G4).  The XOR
// touches byte-address bits 4-7 only, so 16-B and 4-B alignment survive.
__device__ __forceinline__ int swz128(int row, int elem_col) {
    return row * 128 + ((((elem_col) << 1) ^ ((row & 15) << 4)) >> 1);
}

// Same idea on 128-B rows (64 bf16): 8 slots per row, <=2-way residual.
__device__ __forceinline__ int swz64(int row, int elem_col) {
    return row * 64 + ((((elem_col) << 1) ^ ((row & 7) << 4)) >> 1);
}

template <int RS>
__device__ __forceinline__ int swz_rm(int row, int elem_col) {
    return RS == 64 ? swz64(row, elem_col) : swz128(row, elem_col);
}

// Backward-tile indexers.  Measured split (r2 A/B): for D=128 the 256-B
// swizzled rows beat the padded layout (5797->5474us); for D=64 the padded
// odd strides (72/40) are already conflict-free on the fragment reads and
// swizzled power-of-2 rows REGRESSED (1206->1606us) - keep v1's layout.
template <int D> __device__ __forceinline__ int bwd_rm(int row, int col) {
    return D == 64 ? row * 72 + col : swz128(row, col);
}
template <int D> __device__ __forceinline__ int bwd_tr(int row, int col) {
    return D == 64 ? row * 40 + col : swz64(row, col);
}

// Staging-packet decode for the [4][16]-sub-tiled LDS images: one 16-B
// single-row packet per lane with (col-half e, row&3) in the LOW packet
// bits.  The [4][16] tiles are 128 B (bank-aligned), so a b128 store's
// 8-lane group — which under the naive row-pair mapping varies only the
// column packet — hits the same 8 of 32 banks 4 ways (store bank =
// (addr/4)%32; tile stride ≡ 0).  With (e, rw) in the low 3 lane bits the
// 8 addresses land on banks 8*rw + 4*e + [0..3] = all 32 exactly once,
// while each 16-lane (row) subset of a load instruction still covers one
// contiguous 256-B row slice, so global coalescing is unchanged.
template <int D>
__device__ __forceinline__ void pkt_rc(int p, int& row, int& col) {
    constexpr int DLOG = (D / 16 == 8) ? 3 : 2;
    const int rw = (p >> 1) & 3;
    const int dd = (p >> 3) & (D / 16 - 1);
    row = (p >> (3 + DLOG)) * 4 + rw;
    col = dd * 16 + (p & 1) * 8;
}

// C/D layout of mfma_f32_32x32x16_bf16: col = lane&31, row = (r&3) +
// 8*(r>>2) + 4*(lane>>5), r in [0,16).  A/B fragments: i/j = lane&31,
// k = 8*(lane>>5) + elem (verified by mfma_probe32).
template <int D, int WAVES>
__global__ __launch_bounds__(WAVES * WAVE_SIZE)
void attn_fwd_v2_kernel(const bf16* __restrict__ q, const bf16* __restrict__ k,
                        const bf16* __restrict__ v,
                        const float* __restrict__ slopes, float scale,
                        bf16* __restrict__ o, float* __restrict__ lse,
                        int B, int H, int S, int kv_off, int kv_group,
                        int64_t qb, int64_t qh, int64_t qs,
                        int64_t kb, int64_t kh, int64_t ks,
                        int64_t vb, int64_t vh, int64_t vs,
                        int64_t ob, int64_t oh, int64_t os) {
    constexpr int KVB = 64;                 // kv rows per tile
    constexpr int NT = WAVES * WAVE_SIZE;
    constexpr int ROWS = WAVES * 32;        // q rows per workgroup
    constexpr int DCH = D / 16;             // QK^T k-chunks (K=16 per mfma)
    constexpr int DSUB = D / 32;            // O row subtiles (O^T layout)
    constexpr int NPK = KVB * (D / 8);      // K tile 16-B packets
    constexpr int NPV = KVB * (D / 8);      // V tile single-row packets
    constexpr int KPL = (NPK + NT - 1) / NT;
    constexpr int VPL = (NPV + NT - 1) / NT;
    // V image is [kv/4][d/16][kv%4][16]: each [4 kv][16 d] sub-tile is 128 B
    // contiguous, which is exactly what ds_read_b64_tr_b16 gathers (T10) —
    // the PV A-fragment (V^T) comes out of the hardware transpose read, so
    // no scattered u32 transpose-writes (those were 8-way write conflicts =
    // the 61.6% CONF of the r2 PMC) and half the LDS of the [D][128] image.
    constexpr int VST = (D / 16) * 64;      // elements per 4-kv-row group

    __shared__ __attribute__((aligned(16))) bf16 k_lds[2][KVB * 128];
    __shared__ __attribute__((aligned(16))) bf16 vs_lds[2][KVB * D];

    // XCD-affinity remap (T1): the hardware places linear block id L on XCD
    // L%8, so with qblock fastest-varying the q-blocks sharing one (b,h)'s
    // K/V land on 8 different L2s.  When H*B % 8 == 0, re-deriving (qblock,
    // h, b) with (h,b) fastest pins each (b,h) group to ONE XCD, making its
    // K/V tile L2-resident across q-blocks.  Pure speed, never correctness.
    int qblock = blockIdx.x, h = blockIdx.y, b = blockIdx.z;
    {
        const int ngrp = gridDim.y * gridDim.z;  // H * B
        if ((ngrp & 7) == 0) {
            const int flat = blockIdx.x +
                gridDim.x * (blockIdx.y + gridDim.y * blockIdx.z);
            qblock = flat / ngrp;
            const int g = flat % ngrp;
            h = g % gridDim.y;
            b = g / gridDim.y;
        }
    }
    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int l31 = lane & 31;
    const int hi = lane >> 5;

    const int64_t bh_off = ((int64_t)b * H + h) * S;
    const int hk = h / kv_group;
    const bf16* qp = q + b * qb + h * qh;
    const bf16* kp = k + b * kb + hk * kh;
    const bf16* vp = v + b * vb + hk * vh;
    bf16* op = o + b * ob + h * oh;
    // softmax runs in the exp2 domain (v_exp_f32 is natively exp2): fold
    // log2(e) into scale and slope so no per-element multiply is added
    const float LOG2E = 1.4426950408889634f;
    const float scale2 = scale * LOG2E;
    const float slope2 = slopes[h] * LOG2E;

    const int qr0 = qblock * ROWS + wave * 32;
    const int iq = qr0 + l31;

    frag_ab bQ[DCH];
#pragma unroll
    for (int c = 0; c < DCH; ++c) {
        bQ[c] = *reinterpret_cast<const frag_ab*>(
            qp + (int64_t)iq * qs + c * 16 + 8 * hi);
    }

    f32x16 accO[DSUB];
#pragma unroll
    for (int dsb = 0; dsb < DSUB; ++dsb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) accO[dsb][r] = 0.f;
    }
    float m_run = NEG_INF, l_run = 0.f;

    const bool full_vis = (kv_off <= -S);
    int nkv = full_vis ? (S / KVB)
        : (qblock * ROWS + ROWS - kv_off + KVB - 1) / KVB;
    nkv = min(max(nkv, 0), S / KVB);

    frag_ab kreg[KPL];
    frag_ab vreg[VPL];

    auto stage_load = [&](int nb) {
        const int kvrow0 = nb * KVB;
#pragma unroll
        for (int i = 0; i < KPL; ++i) {
            const int p = tid + i * NT;
            if (p < NPK) {
                const int row = p / (D / 8);
                const int col = (p % (D / 8)) * 8;
                kreg[i] = *reinterpret_cast<const frag_ab*>(
                    kp + (int64_t)(kvrow0 + row) * ks + col);
            }
        }
#pragma unroll
        for (int i = 0; i < VPL; ++i) {
            const int p = tid + i * NT;
            if (p < NPV) {
                int row, col;
                pkt_rc<D>(p, row, col);
                vreg[i] = *reinterpret_cast<const frag_ab*>(
                    vp + (int64_t)(kvrow0 + row) * vs + col);
            }
        }
    };
    auto stage_write = [&](int buf) {
#pragma unroll
        for (int i = 0; i < KPL; ++i) {
            const int p = tid + i * NT;
            if (p < NPK) {
                const int row = p / (D / 8);
                const int col = (p % (D / 8)) * 8;
                *reinterpret_cast<frag_ab*>(&k_lds[buf][swz128(row, col)]) =
                    kreg[i];
            }
        }
#pragma unroll
        for (int i = 0; i < VPL; ++i) {
            const int p = tid + i * NT;
            if (p < NPV) {
                int row, col;
                pkt_rc<D>(p, row, col);
                const int idx = (row >> 2) * VST + (col >> 4) * 64 +
                                (row & 3) * 16 + (col & 15);
                *reinterpret_cast<frag_ab*>(&vs_lds[buf][idx]) = vreg[i];
            }
        }
    };

    if (nkv > 0) {
        stage_load(0);
        stage_write(0);
    }
    __syncthreads();

    for (int nb = 0; nb < nkv; ++nb) {
        const int cur = nb & 1;
        const int kvrow0 = nb * KVB;
        const bool last = (nb + 1 == nkv);
        if (!last) stage_load(nb + 1);   // T14 issue-early

        // a tile entirely above this wave's causal diagonal is skipped
        // (wave-uniform branch; staging stays cooperative)
        const bool active = full_vis || (kvrow0 + kv_off <= qr0 + 31);

        f32x16 st[2];
        frag_ab pfrag[4];
        if (active) {
#pragma unroll
            for (int ns = 0; ns < 2; ++ns) {
#pragma unroll
                for (int r = 0; r < 16; ++r) st[ns][r] = 0.f;
            }
#pragma unroll
            for (int ns = 0; ns < 2; ++ns) {
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int c = 0; c < DCH; ++c) {
                    frag_ab aK = *reinterpret_cast<const frag_ab*>(
                        &k_lds[cur][swz128(ns * 32 + l31, c * 16 + 8 * hi)]);
                    st[ns] = MFMA_32x32x16(aK, bQ[c], st[ns]);
                }
                __builtin_amdgcn_s_setprio(0);
            }

            // scale + ALiBi + causal mask (exp2 domain); row max in-register
            // as a log-depth tree (a 32-deep serial fmax chain is ~128
            // dependent cycles)
            float mr[32];
#pragma unroll
            for (int ns = 0; ns < 2; ++ns) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int jk = kvrow0 + ns * 32 + (r & 3) + 8 * (r >> 2) +
                                   4 * hi + kv_off;
                    float x = st[ns][r] * scale2 + slope2 * (float)(jk - iq);
                    x = (jk <= iq) ? x : NEG_INF;
                    st[ns][r] = x;
                    mr[ns * 16 + r] = x;
                }
            }
#pragma unroll
            for (int w = 16; w > 0; w >>= 1) {
#pragma unroll
                for (int i = 0; i < w; ++i) mr[i] = fmaxf(mr[i], mr[i + w]);
            }
            float mx = mr[0];
            {   // lanes l and l+32 hold the same q row's two kv halves
                auto sw = __builtin_amdgcn_permlane32_swap(
                    __float_as_uint(mx), __float_as_uint(mx), false, false);
                mx = fmaxf(u2f(sw[0]), u2f(sw[1]));
            }
            const float m_new = fmaxf(m_run, mx);
            const float alpha = exp2f(m_run - m_new);
            m_run = m_new;
            l_run *= alpha;
#pragma unroll
            for (int dsb = 0; dsb < DSUB; ++dsb) {
#pragma unroll
                for (int r = 0; r < 16; ++r) accO[dsb][r] *= alpha;
            }
            float sr[32];
#pragma unroll
            for (int ns = 0; ns < 2; ++ns) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const float pv = exp2f(st[ns][r] - m_new);
                    st[ns][r] = pv;
                    sr[ns * 16 + r] = pv;
                }
            }
#pragma unroll
            for (int w = 16; w > 0; w >>= 1) {
#pragma unroll
                for (int i = 0; i < w; ++i) sr[i] += sr[i + w];
            }
            float ls = sr[0];
            {
                auto sw = __builtin_amdgcn_permlane32_swap(
                    __float_as_uint(ls), __float_as_uint(ls), false, false);
                ls = u2f(sw[0]) + u2f(sw[1]);
            }
            l_run += ls;

            // pack P into PV B-fragments: lane needs kv = 16*ks + 8*hi + e.
            // Own regs hold kv%8<4 (lower half-wave) / >=4 (upper); one
            // permlane32_swap per word pair completes both fragments (T12).
#pragma unroll
            for (int ns = 0; ns < 2; ++ns) {
#pragma unroll
                for (int hf = 0; hf < 2; ++hf) {
                    union { __bf16 h2[2]; uint32_t u; } wa, wb, wc, wd;
                    wa.h2[0] = (__bf16)st[ns][8 * hf + 0];
                    wa.h2[1] = (__bf16)st[ns][8 * hf + 1];
                    wc.h2[0] = (__bf16)st[ns][8 * hf + 2];
                    wc.h2[1] = (__bf16)st[ns][8 * hf + 3];
                    wb.h2[0] = (__bf16)st[ns][8 * hf + 4];
                    wb.h2[1] = (__bf16)st[ns][8 * hf + 5];
                    wd.h2[0] = (__bf16)st[ns][8 * hf + 6];
                    wd.h2[1] = (__bf16)st[ns][8 * hf + 7];
                    auto s1 = __builtin_amdgcn_permlane32_swap(wa.u, wb.u,
                                                              false, false);
                    auto s2 = __builtin_amdgcn_permlane32_swap(wc.u, wd.u,
                                                              false, false);
                    union { frag_ab f; uint32_t u[4]; } out;
                    out.u[0] = s1[0];
                    out.u[1] = s2[0];
                    out.u[2] = s1[1];
                    out.u[3] = s2[1];
                    pfrag[2 * ns + hf] = out.f;
                }
            }
        }

        if (!last) stage_write(cur ^ 1);  // T14 write-late

        if (active) {
            // O^T += V^T · P^T.  The V^T A-fragment comes from two hardware
            // transpose reads per (dsb, ks): quarter-wave q reads the
            // [4 kv][16 d] sub-tile for d-block 2*dsb+(q&1), kv group
            // 4*ks + 2*hi (+1), and each lane receives its d-column.
            const int lq = lane >> 4;           // quarter-wave index
            const int dblk_off = (lq & 1) * 64; // within-group d-block elems
            const int lel = (lane & 15) * 4;    // lane's element in the tile
#pragma unroll
            for (int dsb = 0; dsb < DSUB; ++dsb) {
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int ks2 = 0; ks2 < 4; ++ks2) {
                    const int g0 = 4 * ks2 + 2 * hi;
                    const int base = g0 * VST + (2 * dsb) * 64 + dblk_off + lel;
                    union { frag_ab f; bf16x4 h[2]; } av;
                    av.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                        (lds_bf16x4_p)&vs_lds[cur][base]);
                    av.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                        (lds_bf16x4_p)&vs_lds[cur][base + VST]);
                    accO[dsb] = MFMA_32x32x16(av.f, pfrag[ks2], accO[dsb]);
                }
                __builtin_amdgcn_s_setprio(0);
            }
        }
        __syncthreads();
    }

    // epilogue: lane owns q row iq; accO reg r is d = 8*(r>>2)+(r&3)+4*hi
    const float inv_l = (l_run > 0.f) ? 1.0f / l_run : 0.f;
    bf16* orow = op + (int64_t)iq * os;
#pragma unroll
    for (int dsb = 0; dsb < DSUB; ++dsb) {
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            union { __bf16 h4[4]; uint2 u; } w;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                w.h4[r] = (__bf16)(accO[dsb][4 * g + r] * inv_l);
            }
            *reinterpret_cast<uint2*>(orow + dsb * 32 + 8 * g + 4 * hi) = w.u;
        }
    }
    if (hi == 0) {
        // stats were kept in the exp2 domain; lse is defined in natural log
        lse[bh_off + iq] = 0.6931471805599453f * (m_run + __log2f(l_run));
    }
}

// semantics probe for ds_read_b64_tr_b16: stage a [4][16] bf16 tile into
// LDS, read it with the transpose instruction the way the PV loop does,
// and write out what each lane received (expected: lane l&15 gets column
// l&15, elements j=0..3 = rows).
__global__ void tr16_probe_kernel(const bf16* __restrict__ in,
                                  float* __restrict__ out) {
    __shared__ __attribute__((aligned(16))) bf16 lds[64];
    const int lane = threadIdx.x;
    if (lane < 8) {
        *reinterpret_cast<frag_ab*>(&lds[lane * 8]) =
            *reinterpret_cast<const frag_ab*>(in + lane * 8);
    }
    __syncthreads();
    if (lane < 16) {
        bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4_p)&lds[lane * 4]);
#pragma unroll
        for (int j = 0; j < 4; ++j) out[lane * 4 + j] = (float)v[j];
    }
}

// layout probe for the 32x32x16 fragment maps (see attn_fwd_v2_kernel)
__global__ void mfma_probe32_kernel(const bf16* __restrict__ A,
                                    const bf16* __restrict__ Bt,
                                    float* __restrict__ C) {
    const int lane = threadIdx.x;
    const int l31 = lane & 31, hi = lane >> 5;
    // A[i][k]: i = l31, k = 8*hi + e (row-major A: 32x16)
    frag_ab a = *reinterpret_cast<const frag_ab*>(A + l31 * 16 + 8 * hi);
    // B[k][j] with Bt stored [j][k] (row-major 32x16): j = l31, k = 8*hi + e
    frag_ab b = *reinterpret_cast<const frag_ab*>(Bt + l31 * 16 + 8 * hi);
    f32x16 c;
#pragma unroll
    for (int r = 0; r < 16; ++r) c[r] = 0.f;
    c = MFMA_32x32x16(a, b, c);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        C[((r & 3) + 8 * (r >> 2) + 4 * hi) * 32 + l31] = c[r];
    }
}

}  // namespace

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor slopes,
                                    double scale, int64_t kv_off) {
    TORCH_CHECK(q.is_cuda() && q.stride(3) == 1 && k.stride(3) == 1 &&
                v.stride(3) == 1, "attn_fwd: last dim must be contiguous");
    TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "attn_fwd: bf16 only");
    const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
    const int Hkv = k.size(1);
    TORCH_CHECK(H % Hkv == 0, "attn_fwd: q heads must be a multiple of kv heads");
    const int kv_group = H / Hkv;
    TORCH_CHECK(S % BLOCK_M == 0, "attn_fwd: S must be a multiple of 64");
    TORCH_CHECK(D == 64 || D == 128, "attn_fwd: head dim 64 or 128");
    TORCH_CHECK(slopes.numel() == H && slopes.scalar_type() == torch::kFloat);

    // O physically [B, S, H, D]: the model reshapes attention output back to
    // [B, S, H*D] for the dense projection — this layout makes that free.
    auto o_phys = torch::empty({B, S, H, D}, q.options());
    auto o = o_phys.permute({0, 2, 1, 3});
    auto lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));

    auto stream = at::cuda::getCurrentCUDAStream();

    // v2 (8-wave 32x32-MFMA ladder kernel) handles S % 128 == 0; v1 covers
    // the rest.  PG_ATTN_V2=0 forces the v1 path for A/B comparisons.
    static const int use_v2 = [] {
        const char* e = getenv("PG_ATTN_V2");
        return e ? atoi(e) : 1;
    }();
    // PG_ATTN_W4=1: 4-wave workgroups (two independent WGs per CU instead
    // of one barrier-locked 8-wave WG — A/B for phase overlap)
    static const int force_w4 = [] {
        const char* e = getenv("PG_ATTN_W4");
        return e ? atoi(e) : 0;
    }();
    if (use_v2 && S % 128 == 0) {
#define LAUNCH_FWD2(DV, WV)                                                   \
    do {                                                                      \
        dim3 grid(S / (32 * WV), H, B);                                       \
        hipLaunchKernelGGL((attn_fwd_v2_kernel<DV, WV>), grid,                \
            dim3(WV * WAVE_SIZE), 0, stream,                                  \
            reinterpret_cast<const bf16*>(q.data_ptr()),                      \
            reinterpret_cast<const bf16*>(k.data_ptr()),                      \
            reinterpret_cast<const bf16*>(v.data_ptr()),                      \
            slopes.data_ptr<float>(), (float)scale,                           \
            reinterpret_cast<bf16*>(o_phys.data_ptr()), lse.data_ptr<float>(),\
            B, H, S, (int)kv_off, kv_group,                                   \
            q.stride(0), q.stride(1), q.stride(2),                            \
            k.stride(0), k.stride(1), k.stride(2),                            \
            v.stride(0), v.stride(1), v.stride(2),                            \
            o.stride(0), o.stride(1), o.stride(2));                           \
    } while (0)
        if (D == 64) {
            if (S % 256 == 0 && !force_w4) LAUNCH_FWD2(64, 8);
            else LAUNCH_FWD2(64, 4);
        } else {
            if (S % 256 == 0 && !force_w4) LAUNCH_FWD2(128, 8);
            else LAUNCH_FWD2(128, 4);
        }
#undef LAUNCH_FWD2
        HIP_CHECK_LAUNCH();
        return {o, lse};
    }

    // MT=2 (128 q rows / workgroup) when S allows: 2x the MFMA work per LDS
    // fragment read
    // D=64: 4 waves x 2 row-tiles.  D=128: default 8 waves x 1 tile
    // (same 128 q rows, lower per-wave registers); PG_ATTN_V128 picks the
    // variant (1: 4w/MT1, 2: 4w/MT2, 3: 8w/MT1).
    static const int v128 = [] {
        const char* e = getenv("PG_ATTN_V128");
        return e ? atoi(e) : 3;  // 8-wave MT1: 181 TF vs 153 (4-wave MT2) measured
    }();
#define LAUNCH_FWD(DV, MTV, WV)                                               \
    do {                                                                      \
        dim3 grid(S / (16 * MTV * WV), H, B);                                 \
        hipLaunchKernelGGL((attn_fwd_kernel<DV, MTV, WV>), grid,              \
            dim3(WV * WAVE_SIZE), 0, stream,                                  \
            reinterpret_cast<const bf16*>(q.data_ptr()),                      \
            reinterpret_cast<const bf16*>(k.data_ptr()),                      \
            reinterpret_cast<const bf16*>(v.data_ptr()),                      \
            slopes.data_ptr<float>(), (float)scale,                           \
            reinterpret_cast<bf16*>(o_phys.data_ptr()), lse.data_ptr<float>(),\
            B, H, S, (int)kv_off, kv_group,                                   \
            q.stride(0), q.stride(1), q.stride(2),                            \
            k.stride(0), k.stride(1), k.stride(2),                            \
            v.stride(0), v.stride(1), v.stride(2),                            \
            o.stride(0), o.stride(1), o.stride(2));                           \
    } while (0)
    static const int v64 = [] {
        const char* e = getenv("PG_ATTN_V64");
        return e ? atoi(e) : 3;  // 8w/MT1: 141 TF vs 135 (4w/MT2) measured
    }();
    if (D == 64) {
        if (S % 128 != 0) LAUNCH_FWD(64, 1, 4);
        else if (v64 == 3) LAUNCH_FWD(64, 1, 8);
        else LAUNCH_FWD(64, 2, 4);
    } else if (S % 128 != 0) {
        LAUNCH_FWD(128, 1, 4);
    } else if (v128 == 3) {
        LAUNCH_FWD(128, 1, 8);
    } else if (v128 == 2) {
        LAUNCH_FWD(128, 2, 4);
    } else {
        LAUNCH_FWD(128, 1, 4);
    }
#undef LAUNCH_FWD
    HIP_CHECK_LAUNCH();
    return {o, lse};
}

torch::Tensor tr16_probe(torch::Tensor tile) {
    TORCH_CHECK(tile.is_cuda() && tile.scalar_type() == torch::kBFloat16 &&
                tile.numel() == 64);
    auto out = torch::empty({16, 4}, tile.options().dtype(torch::kFloat));
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
        reinterpret_cast<const bf16*>(tile.contiguous().data_ptr()),
        out.data_ptr<float>());
    HIP_CHECK_LAUNCH();
    return out;
}

torch::Tensor mfma_probe32(torch::Tensor A, torch::Tensor Bt) {
    TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(A.sizes() == torch::IntArrayRef({32, 16}) &&
                Bt.sizes() == torch::IntArrayRef({32, 16}));
    auto C = torch::empty({32, 32}, A.options().dtype(torch::kFloat));
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, stream,
        reinterpret_cast<const bf16*>(A.contiguous().data_ptr()),
        reinterpret_cast<const bf16*>(Bt.contiguous().data_ptr()),
        C.data_ptr<float>());
    HIP_CHECK_LAUNCH();
    return C;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor Bt) {
    TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}) &&
                Bt.sizes() == torch::IntArrayRef({16, 32}));
    auto C = torch::empty({16, 16}, A.options().dtype(torch::kFloat));
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
        reinterpret_cast<const bf16*>(A.contiguous().data_ptr()),
        reinterpret_cast<const bf16*>(Bt.contiguous().data_ptr()),
        C.data_ptr<float>());
    HIP_CHECK_LAUNCH();
    return C;
}

// ======================================================================
// Backward — two-pass flash2 (no atomics):
//   delta:  delta[m] = rowsum(dO * O)
//   pass A (kv-outer): dK, dV — one workgroup per 64-row kv block,
//                      iterating q blocks >= kv block (causal)
//   pass B (q-outer):  dQ — one workgroup per 64-row q block,
//                      iterating kv blocks <= q block
// Both passes recompute S and P from (q, k, lse): MFMA FLOPs are cheap
// next to re-reading P from HBM.
namespace {

template <int D>
__global__ __launch_bounds__(256)
void attn_bwd_delta_kernel(const bf16* __restrict__ dout,
                           const bf16* __restrict__ o,
                           float* __restrict__ delta, int64_t rows,
                           int H, int S,
                           int64_t db, int64_t dh, int64_t ds_,
                           int64_t ob2, int64_t oh2, int64_t os2) {
    const int64_t row = (int64_t)blockIdx.x * 4 + threadIdx.x / WAVE_SIZE;
    if (row >= rows) return;
    const int lane = threadIdx.x % WAVE_SIZE;
    const int64_t b = row / ((int64_t)H * S);
    const int64_t h = (row / S) % H;
    const int64_t s = row % S;
    const bf16* dr = dout + b * db + h * dh + s * ds_;
    const bf16* orow = o + b * ob2 + h * oh2 + s * os2;
    float acc = 0.f;
    for (int i = lane; i < D; i += WAVE_SIZE) {
        acc += to_float(dr[i]) * to_float(orow[i]);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) delta[row] = acc;
}

// pass A: dK(n,:) = scale * sum_m dS(m,n) Q(m,:);  dV(n,:) = sum_m P(m,n) dO(m,:)
// QR = q rows staged per iteration: 32 keeps LDS under 1/3 of the 160 KB CU
// budget (at QR=64 the kernel was LDS-capped to ONE wave/SIMD).
// W = waves per workgroup: 8 waves own 128 kv rows and SHARE each staged
// q-chunk, halving the Q/dO staging traffic vs 4-wave blocks.
template <int D, int QR, int W>
__global__ __launch_bounds__(W * WAVE_SIZE)
void attn_bwd_dkdv_kernel(const bf16* __restrict__ dout,
                          const bf16* __restrict__ q, const bf16* __restrict__ k,
                          const bf16* __restrict__ v,
                          const float* __restrict__ lse,
                          const float* __restrict__ delta,
                          const float* __restrict__ slopes, float scale,
                          bf16* __restrict__ dk, bf16* __restrict__ dv,
                          int B, int H, int S, int kv_off, int kv_group,
                          int64_t gb, int64_t gh, int64_t gs,
                          int64_t qb2, int64_t qh2, int64_t qs2,
                          int64_t kb2, int64_t kh2, int64_t ks2,
                          int64_t vb2, int64_t vh2, int64_t vs2,
                          int64_t wb, int64_t wh, int64_t ws) {
    constexpr int DCH = D / 32;
    constexpr int DSUB = D / 16;
    constexpr int NT = W * WAVE_SIZE;
    constexpr int KVROWS = W * 16;     // kv rows per workgroup

    // round-2 retrofit: power-of-2 row strides with the T2 XOR swizzle
    // (row-major tiles on 256-B rows via swz128 for D=128 — 128-B rows via
    // swz64 for D=64, keeping the LDS footprint and occupancy of r1 —
    // transposed tiles on 128-B rows via swz64) replace the +8-element
    // padding: r1 PMC measured 33-63% of LDS cycles in bank conflicts.
    constexpr int RMS = (D == 64) ? 72 : 128;   // row-major tile stride
    constexpr int TRS = (D == 64) ? 40 : 64;    // transposed tile stride
    __shared__ bf16 do_lds[QR * RMS];
    __shared__ bf16 q_lds[QR * RMS];
    __shared__ bf16 qt_lds[D * TRS];
    __shared__ bf16 dot_lds[D * TRS];
    __shared__ bf16 pt_lds[W][16 * TRS];   // P^T  (n rows, m cols)
    __shared__ bf16 dst_lds[W][16 * TRS];  // dS^T (n rows, m cols)
    __shared__ float lse_lds[QR];
    __shared__ float delta_lds[QR];

    const int nb = blockIdx.x;
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lgrp = lane >> 4, lcol = lane & 15;

    // blockIdx.y is the KV head; the group's q heads accumulate into the
    // same dk/dv (GQA — looping here avoids cross-block write races)
    const int kv0 = nb * KVROWS + wave * 16;  // this wave's 16 kv rows

    // this wave's K and V rows live in registers (A-fragments over d)
    frag_ab aK[DCH], aV[DCH];
    {
        const bf16* krow = k + b * kb2 + h * kh2 + (int64_t)(kv0 + lcol) * ks2;
        const bf16* vrow = v + b * vb2 + h * vh2 + (int64_t)(kv0 + lcol) * vs2;
#pragma unroll
        for (int c = 0; c < DCH; ++c) {
            aK[c] = *reinterpret_cast<const frag_ab*>(krow + c * 32 + 8 * lgrp);
            aV[c] = *reinterpret_cast<const frag_ab*>(vrow + c * 32 + 8 * lgrp);
        }
    }

    frag_cd accDK[DSUB], accDV[DSUB];
#pragma unroll
    for (int s = 0; s < DSUB; ++s) {
        accDK[s] = frag_cd{0.f, 0.f, 0.f, 0.f};
        accDV[s] = frag_cd{0.f, 0.f, 0.f, 0.f};
    }

    for (int gq = 0; gq < kv_group; ++gq) {
    const int hq = h * kv_group + gq;
    const int64_t bh_off = ((int64_t)b * H + hq) * S;  // lse/delta layout
    const bf16* qp = q + b * qb2 + hq * qh2;
    const bf16* dop = dout + b * gb + hq * gh;
    const float slope = slopes[hq];

    const float LOG2E = 1.4426950408889634f;
    const float scale2 = scale * LOG2E;
    const float slope2 = slope * LOG2E;
    const int q_start = (kv_off <= -S) ? 0 : nb * KVROWS + kv_off;
    for (int q0 = q_start < 0 ? 0 : q_start; q0 < S; q0 += QR) {
        __syncthreads();
        {   // stage Q and dO, row-major + transposed, plus lse/delta
            constexpr int PACKETS = (QR / 2) * (D / 8);
            for (int p = tid; p < PACKETS; p += NT) {
                const int row = (p / (D / 8)) * 2;
                const int col = (p % (D / 8)) * 8;
                frag_ab pq0 = *reinterpret_cast<const frag_ab*>(
                    qp + (int64_t)(q0 + row) * qs2 + col);
                frag_ab pq1 = *reinterpret_cast<const frag_ab*>(
                    qp + (int64_t)(q0 + row + 1) * qs2 + col);
                frag_ab pd0 = *reinterpret_cast<const frag_ab*>(
                    dop + (int64_t)(q0 + row) * gs + col);
                frag_ab pd1 = *reinterpret_cast<const frag_ab*>(
                    dop + (int64_t)(q0 + row + 1) * gs + col);
                *reinterpret_cast<frag_ab*>(&q_lds[bwd_rm<D>(row, col)]) = pq0;
                *reinterpret_cast<frag_ab*>(&q_lds[bwd_rm<D>(row + 1, col)]) = pq1;
                *reinterpret_cast<frag_ab*>(&do_lds[bwd_rm<D>(row, col)]) = pd0;
                *reinterpret_cast<frag_ab*>(&do_lds[bwd_rm<D>(row + 1, col)]) = pd1;
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    union { __bf16 h[2]; uint32_t u; } a, b2;
                    a.h[0] = pq0[j]; a.h[1] = pq1[j];
                    b2.h[0] = pd0[j]; b2.h[1] = pd1[j];
                    *reinterpret_cast<uint32_t*>(&qt_lds[bwd_tr<D>(col + j, row)]) = a.u;
                    *reinterpret_cast<uint32_t*>(&dot_lds[bwd_tr<D>(col + j, row)]) = b2.u;
                }
            }
            for (int i = tid; i < QR; i += NT) {
                lse_lds[i] = lse[bh_off + q0 + i] * LOG2E;
                delta_lds[i] = delta[bh_off + q0 + i];
            }
        }
        __syncthreads();

        // build P^T and dS^T for ALL QR q columns of this chunk
#pragma unroll
        for (int ms = 0; ms < QR / 16; ++ms) {
            frag_cd st = frag_cd{0.f, 0.f, 0.f, 0.f};
            frag_cd dpt = frag_cd{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int c = 0; c < DCH; ++c) {
                frag_ab bQ = *reinterpret_cast<const frag_ab*>(
                    &q_lds[bwd_rm<D>(ms * 16 + lcol, c * 32 + 8 * lgrp)]);
                st = MFMA_16x16x32(aK[c], bQ, st);          // S^T = K·Q^T
                frag_ab bDO = *reinterpret_cast<const frag_ab*>(
                    &do_lds[bwd_rm<D>(ms * 16 + lcol, c * 32 + 8 * lgrp)]);
                dpt = MFMA_16x16x32(aV[c], bDO, dpt);       // dP^T = V·dO^T
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int n_glob = kv0 + 4 * lgrp + r + kv_off;
                const int m_loc = ms * 16 + lcol;
                const int m_glob = q0 + m_loc;
                float z = st[r] * scale2 + slope2 * (float)(n_glob - m_glob);
                float p = (n_glob <= m_glob) ? exp2f(z - lse_lds[m_loc]) : 0.f;
                float ds = p * (dpt[r] - delta_lds[m_loc]) * scale;
                pt_lds[wave][bwd_tr<D>(4 * lgrp + r, m_loc)] = (bf16)p;
                dst_lds[wave][bwd_tr<D>(4 * lgrp + r, m_loc)] = (bf16)ds;
            }
        }
        __builtin_amdgcn_s_waitcnt(0);  // own-wave LDS writes visible

        // dV += P^T·dO and dK += dS^T·Q over the QR q rows
#pragma unroll
        for (int kc = 0; kc < QR / 32; ++kc) {
            frag_ab aPT = *reinterpret_cast<const frag_ab*>(
                &pt_lds[wave][bwd_tr<D>(lcol, kc * 32 + 8 * lgrp)]);
            frag_ab aDST = *reinterpret_cast<const frag_ab*>(
                &dst_lds[wave][bwd_tr<D>(lcol, kc * 32 + 8 * lgrp)]);
#pragma unroll
            for (int ds = 0; ds < DSUB; ++ds) {
                frag_ab bDOT = *reinterpret_cast<const frag_ab*>(
                    &dot_lds[bwd_tr<D>(ds * 16 + lcol, kc * 32 + 8 * lgrp)]);
                accDV[ds] = MFMA_16x16x32(aPT, bDOT, accDV[ds]);
                frag_ab bQT = *reinterpret_cast<const frag_ab*>(
                    &qt_lds[bwd_tr<D>(ds * 16 + lcol, kc * 32 + 8 * lgrp)]);
                accDK[ds] = MFMA_16x16x32(aDST, bQT, accDK[ds]);
            }
        }
    }

    }  // gq (GQA group loop)

    // epilogue: C row = this wave's kv row (4*lgrp+reg), col = d
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int n_glob = kv0 + 4 * lgrp + r;
        bf16* dkrow = dk + b * wb + h * wh + (int64_t)n_glob * ws;
        bf16* dvrow = dv + b * wb + h * wh + (int64_t)n_glob * ws;
#pragma unroll
        for (int ds = 0; ds < DSUB; ++ds) {
            dkrow[ds * 16 + lcol] = (bf16)accDK[ds][r];
            dvrow[ds * 16 + lcol] = (bf16)accDV[ds][r];
        }
    }
}

// pass B: dQ(m,:) = scale * sum_n dS(m,n) K(n,:)
// KVR = kv rows staged per iteration (same LDS-occupancy reasoning as QR);
// W waves (8 when S allows) share each staged kv chunk.
template <int D, int KVR, int W>
__global__ __launch_bounds__(W * WAVE_SIZE)
void attn_bwd_dq_kernel(const bf16* __restrict__ dout,
                        const bf16* __restrict__ q, const bf16* __restrict__ k,
                        const bf16* __restrict__ v,
                        const float* __restrict__ lse,
                        const float* __restrict__ delta,
                        const float* __restrict__ slopes, float scale,
                        bf16* __restrict__ dq,
                        int B, int H, int S, int kv_off, int kv_group,
                        int64_t gb, int64_t gh, int64_t gs,
                        int64_t qb2, int64_t qh2, int64_t qs2,
                        int64_t kb2, int64_t kh2, int64_t ks2,
                        int64_t vb2, int64_t vh2, int64_t vs2,
                        int64_t wb, int64_t wh, int64_t ws) {
    constexpr int DCH = D / 32;
    constexpr int DSUB = D / 16;
    constexpr int NT = W * WAVE_SIZE;
    constexpr int QROWS = W * 16;          // q rows per workgroup

    // swizzled strides, same scheme as dkdv (see comment there)
    constexpr int RMS = (D == 64) ? 72 : 128;
    constexpr int TRS = (D == 64) ? 40 : 64;
    __shared__ bf16 k_lds[KVR * RMS];       // row-major K (QK^T B-frags)
    __shared__ bf16 v_lds[KVR * RMS];       // row-major V (dP B-frags)
    __shared__ bf16 kt_lds[D * TRS];        // transposed K (dQ B-frags)
    __shared__ bf16 ds_lds[W][16 * TRS];    // dS (m rows, n cols)

    const int qb = blockIdx.x;
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lgrp = lane >> 4, lcol = lane & 15;

    const int64_t bh_off = ((int64_t)b * H + h) * S;  // lse/delta layout
    const int hk = h / kv_group;  // GQA
    const bf16* kp = k + b * kb2 + hk * kh2;
    const bf16* vp = v + b * vb2 + hk * vh2;
    const float slope = slopes[h];

    const int qrow0 = qb * QROWS + wave * 16;

    // Q and dO rows of this wave in registers; lse/delta per owned row
    frag_ab aQ[DCH], aDO[DCH];
    {
        const bf16* qrow = q + b * qb2 + h * qh2 + (int64_t)(qrow0 + lcol) * qs2;
        const bf16* dorow = dout + b * gb + h * gh + (int64_t)(qrow0 + lcol) * gs;
#pragma unroll
        for (int c = 0; c < DCH; ++c) {
            aQ[c] = *reinterpret_cast<const frag_ab*>(qrow + c * 32 + 8 * lgrp);
            aDO[c] = *reinterpret_cast<const frag_ab*>(dorow + c * 32 + 8 * lgrp);
        }
    }
    const float LOG2E = 1.4426950408889634f;
    const float scale2 = scale * LOG2E;
    const float slope2 = slope * LOG2E;
    float lse_r[4], delta_r[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int iq = qrow0 + 4 * lgrp + r;
        lse_r[r] = lse[bh_off + iq] * LOG2E;
        delta_r[r] = delta[bh_off + iq];
    }

    frag_cd accDQ[DSUB];
#pragma unroll
    for (int s = 0; s < DSUB; ++s) accDQ[s] = frag_cd{0.f, 0.f, 0.f, 0.f};

    const int kv_end = (kv_off <= -S) ? S
        : min((int64_t)S, (int64_t)qb * QROWS + QROWS - kv_off);  // causal bound
    for (int kv0 = 0; kv0 < kv_end; kv0 += KVR) {
        __syncthreads();
        {   // stage K (row-major + transposed) and V (row-major)
            constexpr int PACKETS = (KVR / 2) * (D / 8);
            for (int p = tid; p < PACKETS; p += NT) {
                const int row = (p / (D / 8)) * 2;
                const int col = (p % (D / 8)) * 8;
                frag_ab pk0 = *reinterpret_cast<const frag_ab*>(
                    kp + (int64_t)(kv0 + row) * ks2 + col);
                frag_ab pk1 = *reinterpret_cast<const frag_ab*>(
                    kp + (int64_t)(kv0 + row + 1) * ks2 + col);
                frag_ab pv0 = *reinterpret_cast<const frag_ab*>(
                    vp + (int64_t)(kv0 + row) * vs2 + col);
                frag_ab pv1 = *reinterpret_cast<const frag_ab*>(
                    vp + (int64_t)(kv0 + row + 1) * vs2 + col);
                *reinterpret_cast<frag_ab*>(&k_lds[bwd_rm<D>(row, col)]) = pk0;
                *reinterpret_cast<frag_ab*>(&k_lds[bwd_rm<D>(row + 1, col)]) = pk1;
                *reinterpret_cast<frag_ab*>(&v_lds[bwd_rm<D>(row, col)]) = pv0;
                *reinterpret_cast<frag_ab*>(&v_lds[bwd_rm<D>(row + 1, col)]) = pv1;
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    union { __bf16 h[2]; uint32_t u; } a;
                    a.h[0] = pk0[j]; a.h[1] = pk1[j];
                    *reinterpret_cast<uint32_t*>(&kt_lds[bwd_tr<D>(col + j, row)]) = a.u;
                }
            }
        }
        __syncthreads();

        // dS for all KVR kv cols of this chunk
#pragma unroll
        for (int ns = 0; ns < KVR / 16; ++ns) {
            frag_cd sacc = frag_cd{0.f, 0.f, 0.f, 0.f};
            frag_cd dpacc = frag_cd{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int c = 0; c < DCH; ++c) {
                frag_ab bK = *reinterpret_cast<const frag_ab*>(
                    &k_lds[bwd_rm<D>(ns * 16 + lcol, c * 32 + 8 * lgrp)]);
                sacc = MFMA_16x16x32(aQ[c], bK, sacc);      // S = Q·K^T
                frag_ab bV = *reinterpret_cast<const frag_ab*>(
                    &v_lds[bwd_rm<D>(ns * 16 + lcol, c * 32 + 8 * lgrp)]);
                dpacc = MFMA_16x16x32(aDO[c], bV, dpacc);   // dP = dO·V^T
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int iq = qrow0 + 4 * lgrp + r;
                const int jk = kv0 + ns * 16 + lcol + kv_off;
                float z = sacc[r] * scale2 + slope2 * (float)(jk - iq);
                float p = (jk <= iq) ? exp2f(z - lse_r[r]) : 0.f;
                float ds = p * (dpacc[r] - delta_r[r]) * scale;
                // store transposed to [m][n] so the dQ A-frag read is linear
                ds_lds[wave][bwd_tr<D>(4 * lgrp + r, ns * 16 + lcol)] = (bf16)ds;
            }
        }
        __builtin_amdgcn_s_waitcnt(0);

        // dQ += dS·K over the KVR kv rows
#pragma unroll
        for (int kc = 0; kc < KVR / 32; ++kc) {
            frag_ab aDS = *reinterpret_cast<const frag_ab*>(
                &ds_lds[wave][bwd_tr<D>(lcol, kc * 32 + 8 * lgrp)]);
#pragma unroll
            for (int ds = 0; ds < DSUB; ++ds) {
                frag_ab bKT = *reinterpret_cast<const frag_ab*>(
                    &kt_lds[bwd_tr<D>(ds * 16 + lcol, kc * 32 + 8 * lgrp)]);
                accDQ[ds] = MFMA_16x16x32(aDS, bKT, accDQ[ds]);
            }
        }
    }

#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int iq = qrow0 + 4 * lgrp + r;
        bf16* dqrow = dq + b * wb + h * wh + (int64_t)iq * ws;
#pragma unroll
        for (int ds = 0; ds < DSUB; ++ds) {
            dqrow[ds * 16 + lcol] = (bf16)accDQ[ds][r];
        }
    }
}

}  // namespace

// ======================================================================
// Backward v2 — same ladder as the forward (32x32 MFMA, lane-local
// layouts, swizzled/subtiled LDS, exp2 domain).  Two kernels:
//   dq_v2   (q-outer):  wave owns 32 q rows; K/V tiles staged; St^T and
//           dP^T have q on the lane axis (lse/delta are per-lane scalars),
//           dS packs into B-fragments and dQ^T accumulates lane-locally.
//   dkdv_v2 (kv-outer): wave owns 32 kv rows (K/V B-fragments live in
//           registers); Q/dO tiles staged in the sub-tiled image, which
//           serves BOTH the row-major A-fragments (St, dP) and the
//           hardware-transpose reads (dV^T = dO^T·P, dK^T = Q^T·dS).
// The sub-tiled image is [rows/4][d/16][4][16] with the d-block position
// XORed by the row-group (tiles stay 128-B contiguous for tr16 while the
// row-major reads spread across banks).
namespace {

// element index into a sub-tiled [R][D] image; DBLK = D/16
template <int D>
__device__ __forceinline__ int sub_idx(int row, int elem_col) {
    constexpr int DBLK = D / 16;
    const int g = row >> 2;
    const int dblk = (elem_col >> 4) ^ (g & (DBLK - 1));
    return g * (DBLK * 64) + dblk * 64 + (row & 3) * 16 + (elem_col & 15);
}

// byte-base of the [4][16] tr16 tile for (row group g, logical d-block)
template <int D>
__device__ __forceinline__ int sub_tile_base(int g, int dblk_log) {
    constexpr int DBLK = D / 16;
    return g * (DBLK * 64) + (dblk_log ^ (g & (DBLK - 1))) * 64;
}

template <int D, int WAVES>
__global__ __launch_bounds__(WAVES * WAVE_SIZE)
void attn_bwd_dq_v2_kernel(const bf16* __restrict__ dout,
                           const bf16* __restrict__ q, const bf16* __restrict__ k,
                           const bf16* __restrict__ v,
                           const float* __restrict__ lse,
                           const float* __restrict__ delta,
                           const float* __restrict__ slopes, float scale,
                           bf16* __restrict__ dq,
                           int B, int H, int S, int kv_off, int kv_group,
                           int64_t gb, int64_t gh, int64_t gs,
                           int64_t qb2, int64_t qh2, int64_t qs2,
                           int64_t kb2, int64_t kh2, int64_t ks2,
                           int64_t vb2, int64_t vh2, int64_t vs2,
                           int64_t wb, int64_t wh, int64_t ws) {
    constexpr int KVB = 64;
    constexpr int NT = WAVES * WAVE_SIZE;
    constexpr int ROWS = WAVES * 32;
    constexpr int DCH = D / 16;
    constexpr int DSUB = D / 32;
    constexpr int VST = (D / 16) * 64;
    constexpr int NPK = KVB * (D / 8);          // K single-row packets
    constexpr int KPL = (NPK + NT - 1) / NT;
    constexpr int NP = (KVB / 2) * (D / 8);     // V row-pair packets
    constexpr int PPL = (NP + NT - 1) / NT;

    __shared__ __attribute__((aligned(16))) bf16 ks_lds[2][KVB * D];
    __shared__ __attribute__((aligned(16))) bf16 v_lds[2][KVB * 128];

    const int qblock = blockIdx.x;
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int l31 = lane & 31;
    const int hi = lane >> 5;
    const int lq = lane >> 4;

    const int64_t bh_off = ((int64_t)b * H + h) * S;
    const int hk = h / kv_group;
    const bf16* kp = k + b * kb2 + hk * kh2;
    const bf16* vp = v + b * vb2 + hk * vh2;
    const float LOG2E = 1.4426950408889634f;
    const float scale2 = scale * LOG2E;
    const float slope2 = slopes[h] * LOG2E;

    const int qr0 = qblock * ROWS + wave * 32;
    const int iq = qr0 + l31;

    frag_ab bQ[DCH], bDO[DCH];
    {
        const bf16* qrow = q + b * qb2 + h * qh2 + (int64_t)iq * qs2;
        const bf16* dorow = dout + b * gb + h * gh + (int64_t)iq * gs;
#pragma unroll
        for (int c = 0; c < DCH; ++c) {
            bQ[c] = *reinterpret_cast<const frag_ab*>(qrow + c * 16 + 8 * hi);
            bDO[c] = *reinterpret_cast<const frag_ab*>(dorow + c * 16 + 8 * hi);
        }
    }
    const float lse2 = lse[bh_off + iq] * LOG2E;
    const float dlt = delta[bh_off + iq];

    f32x16 accDQ[DSUB];
#pragma unroll
    for (int dsb = 0; dsb < DSUB; ++dsb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) accDQ[dsb][r] = 0.f;
    }

    const bool full_vis = (kv_off <= -S);
    const int kv_end = full_vis ? S
        : min(S, qblock * ROWS + ROWS - kv_off);
    const int nkv = max(0, (kv_end + KVB - 1) / KVB);

    frag_ab kreg[KPL], vreg[PPL][2];
    auto stage_load = [&](int nb) {
        const int kvrow0 = nb * KVB;
#pragma unroll
        for (int i = 0; i < KPL; ++i) {
            const int p = tid + i * NT;
            if (p < NPK) {
                int row, col;
                pkt_rc<D>(p, row, col);
                kreg[i] = *reinterpret_cast<const frag_ab*>(
                    kp + (int64_t)(kvrow0 + row) * ks2 + col);
            }
        }
#pragma unroll
        for (int i = 0; i < PPL; ++i) {
            const int p = tid + i * NT;
            if (p < NP) {
                const int row = (p / (D / 8)) * 2;
                const int col = (p % (D / 8)) * 8;
                vreg[i][0] = *reinterpret_cast<const frag_ab*>(
                    vp + (int64_t)(kvrow0 + row) * vs2 + col);
                vreg[i][1] = *reinterpret_cast<const frag_ab*>(
                    vp + (int64_t)(kvrow0 + row + 1) * vs2 + col);
            }
        }
    };
    auto stage_write = [&](int buf) {
#pragma unroll
        for (int i = 0; i < KPL; ++i) {
            const int p = tid + i * NT;
            if (p < NPK) {
                int row, col;
                pkt_rc<D>(p, row, col);
                *reinterpret_cast<frag_ab*>(
                    &ks_lds[buf][sub_idx<D>(row, col)]) = kreg[i];
            }
        }
#pragma unroll
        for (int i = 0; i < PPL; ++i) {
            const int p = tid + i * NT;
            if (p < NP) {
                const int row = (p / (D / 8)) * 2;
                const int col = (p % (D / 8)) * 8;
#pragma unroll
                for (int rr = 0; rr < 2; ++rr) {
                    *reinterpret_cast<frag_ab*>(
                        &v_lds[buf][swz128(row + rr, col)]) = vreg[i][rr];
                }
            }
        }
    };

    constexpr bool EARLY_LOAD = true;
    if (nkv > 0) {
        stage_load(0);
        stage_write(0);
    }
    __syncthreads();

    for (int nb = 0; nb < nkv; ++nb) {
        const int cur = nb & 1;
        const int kvrow0 = nb * KVB;
        const bool last = (nb + 1 == nkv);
        if (EARLY_LOAD && !last) stage_load(nb + 1);

        const bool active = full_vis || (kvrow0 + kv_off <= qr0 + 31);
        if (active) {
            // D=128 must NOT unroll (doubled live f32x16 pairs spill);
            // D=64 has register headroom and full unroll measured faster
            auto ns_body = [&](int ns) {
                frag_ab dsfrag[2];
                f32x16 st, dp;
#pragma unroll
                for (int r = 0; r < 16; ++r) { st[r] = 0.f; dp[r] = 0.f; }
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int c = 0; c < DCH; ++c) {
                    // St^T = K·Q^T: A = K row-major from the sub-tiled image
                    frag_ab aK = *reinterpret_cast<const frag_ab*>(
                        &ks_lds[cur][sub_idx<D>(ns * 32 + l31, c * 16 + 8 * hi)]);
                    st = MFMA_32x32x16(aK, bQ[c], st);
                    frag_ab aV = *reinterpret_cast<const frag_ab*>(
                        &v_lds[cur][swz128(ns * 32 + l31, c * 16 + 8 * hi)]);
                    dp = MFMA_32x32x16(aV, bDO[c], dp);  // dP^T = V·dO^T
                }
                __builtin_amdgcn_s_setprio(0);
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int jk = kvrow0 + ns * 32 + (r & 3) + 8 * (r >> 2) +
                                   4 * hi + kv_off;
                    const float z2 = st[r] * scale2 +
                                     slope2 * (float)(jk - iq);
                    const float p = (jk <= iq) ? exp2f(z2 - lse2) : 0.f;
                    st[r] = p * (dp[r] - dlt) * scale;  // dS^T
                }
                // pack dS^T into B-fragments (k = kv), as in the forward
#pragma unroll
                for (int hf = 0; hf < 2; ++hf) {
                    union { __bf16 h2[2]; uint32_t u; } wa, wb2, wc, wd;
                    wa.h2[0] = (__bf16)st[8 * hf + 0];
                    wa.h2[1] = (__bf16)st[8 * hf + 1];
                    wc.h2[0] = (__bf16)st[8 * hf + 2];
                    wc.h2[1] = (__bf16)st[8 * hf + 3];
                    wb2.h2[0] = (__bf16)st[8 * hf + 4];
                    wb2.h2[1] = (__bf16)st[8 * hf + 5];
                    wd.h2[0] = (__bf16)st[8 * hf + 6];
                    wd.h2[1] = (__bf16)st[8 * hf + 7];
                    auto s1 = __builtin_amdgcn_permlane32_swap(wa.u, wb2.u,
                                                              false, false);
                    auto s2 = __builtin_amdgcn_permlane32_swap(wc.u, wd.u,
                                                              false, false);
                    union { frag_ab f; uint32_t u[4]; } out;
                    out.u[0] = s1[0];
                    out.u[1] = s2[0];
                    out.u[2] = s1[1];
                    out.u[3] = s2[1];
                    dsfrag[hf] = out.f;
                }
                // dQ^T += K^T·dS^T for this ns's two kv chunks (keeps the
                // fragment set at 2 so D=128 stays under the VGPR cap)
#pragma unroll
                for (int dsb = 0; dsb < DSUB; ++dsb) {
                    __builtin_amdgcn_s_setprio(1);
#pragma unroll
                    for (int ksq = 0; ksq < 2; ++ksq) {
                        const int g0 = 8 * ns + 4 * ksq + 2 * hi;
                        const int dblk = 2 * dsb + (lq & 1);
                        union { frag_ab f; bf16x4 h[2]; } akt;
                        akt.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                            (lds_bf16x4_p)&ks_lds[cur][
                                sub_tile_base<D>(g0, dblk) + (lane & 15) * 4]);
                        akt.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                            (lds_bf16x4_p)&ks_lds[cur][
                                sub_tile_base<D>(g0 + 1, dblk) + (lane & 15) * 4]);
                        accDQ[dsb] = MFMA_32x32x16(akt.f, dsfrag[ksq],
                                                   accDQ[dsb]);
                    }
                    __builtin_amdgcn_s_setprio(0);
                }
            };
            if (D == 128) {
#pragma unroll 1
                for (int ns = 0; ns < 2; ++ns) ns_body(ns);
            } else {
                ns_body(0);
                ns_body(1);
            }
        }
        if (!last) {
            if (!EARLY_LOAD) stage_load(nb + 1);
            stage_write(cur ^ 1);
        }
        __syncthreads();
    }

    // epilogue: dq row iq; reg r is d = 8*(r>>2)+(r&3)+4*hi  (dS already
    // carries the softmax-scale factor)
    bf16* dqrow = dq + b * wb + h * wh + (int64_t)iq * ws;
#pragma unroll
    for (int dsb = 0; dsb < DSUB; ++dsb) {
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            union { __bf16 h4[4]; uint2 u; } w;
#pragma unroll
            for (int r = 0; r < 4; ++r) w.h4[r] = (__bf16)accDQ[dsb][4 * g + r];
            *reinterpret_cast<uint2*>(dqrow + dsb * 32 + 8 * g + 4 * hi) = w.u;
        }
    }
}

// DKPART: -1 = accumulate dK over all of D; 0/1 = only that d-half (the
// dk pass at D=128 exceeds the 8-wave register cap with a full-width
// accumulator — two half-width launches recompute St/dP but run at
// 2 waves/SIMD instead of the 4-wave build's 1)
template <int D, int WAVES, bool DO_DK, bool DO_DV, int DKPART = -1>
__global__ __launch_bounds__(WAVES * WAVE_SIZE)
void attn_bwd_dkdv_v2_kernel(const bf16* __restrict__ dout,
                             const bf16* __restrict__ q, const bf16* __restrict__ k,
                             const bf16* __restrict__ v,
                             const float* __restrict__ lse,
                             const float* __restrict__ delta,
                             const float* __restrict__ slopes, float scale,
                             bf16* __restrict__ dk, bf16* __restrict__ dv,
                             int B, int H, int S, int kv_off, int kv_group,
                             int64_t gb, int64_t gh, int64_t gs,
                             int64_t qb2, int64_t qh2, int64_t qs2,
                             int64_t kb2, int64_t kh2, int64_t ks2,
                             int64_t vb2, int64_t vh2, int64_t vs2,
                             int64_t wb, int64_t wh, int64_t ws) {
    constexpr int QT = 64;                   // q rows staged per tile
    constexpr int NT = WAVES * WAVE_SIZE;
    constexpr int ROWS = WAVES * 32;         // kv rows per workgroup
    constexpr int DCH = D / 16;
    constexpr int DSUB = D / 32;
    constexpr int NP = QT * (D / 8);         // single-row staging packets
    constexpr int PPL = (NP + NT - 1) / NT;

    __shared__ __attribute__((aligned(16))) bf16 qs_lds[2][QT * D];
    __shared__ __attribute__((aligned(16))) bf16 dos_lds[2][QT * D];
    __shared__ float lse_lds[2][QT];
    __shared__ float dlt_lds[2][QT];

    const int nb = blockIdx.x;
    const int h = blockIdx.y;   // KV head
    const int b = blockIdx.z;
    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int l31 = lane & 31;
    const int hi = lane >> 5;
    const int lq = lane >> 4;

    const int kv0w = nb * ROWS + wave * 32;  // wave's kv rows
    const int n_glob = kv0w + l31;
    const float LOG2E = 1.4426950408889634f;
    const float scale2 = scale * LOG2E;

    constexpr int DK0 = (DKPART < 0) ? 0 : DKPART * (DSUB / 2);
    constexpr int NDSB = (DKPART < 0) ? DSUB : DSUB / 2;
    frag_ab bK[DCH], bV[DO_DK ? DCH : 1];
    {
        const bf16* krow = k + b * kb2 + h * kh2 + (int64_t)n_glob * ks2;
        const bf16* vrow = v + b * vb2 + h * vh2 + (int64_t)n_glob * vs2;
#pragma unroll
        for (int c = 0; c < DCH; ++c) {
            bK[c] = *reinterpret_cast<const frag_ab*>(krow + c * 16 + 8 * hi);
            if (DO_DK)
                bV[c] = *reinterpret_cast<const frag_ab*>(vrow + c * 16 + 8 * hi);
        }
    }
    f32x16 accDK[DO_DK ? NDSB : 1], accDV[DO_DV ? DSUB : 1];
#pragma unroll
    for (int dsb = 0; dsb < DSUB; ++dsb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
            if (DO_DK && dsb < NDSB) accDK[dsb][r] = 0.f;
            if (DO_DV) accDV[dsb][r] = 0.f;
        }
    }

    const bool full_vis = (kv_off <= -S);
    const int q_first = full_vis ? 0
        : max(0, (nb * ROWS + kv_off) & ~(QT - 1));

    for (int gq = 0; gq < kv_group; ++gq) {
        const int hq = h * kv_group + gq;
        const int64_t bh_off = ((int64_t)b * H + hq) * S;
        const bf16* qp = q + b * qb2 + hq * qh2;
        const bf16* dop = dout + b * gb + hq * gh;
        const float slope2 = slopes[hq] * LOG2E;

        frag_ab qreg[PPL], doreg[PPL];
        auto stage_load = [&](int q0) {
#pragma unroll
            for (int i = 0; i < PPL; ++i) {
                const int p = tid + i * NT;
                if (p < NP) {
                    int row, col;
                    pkt_rc<D>(p, row, col);
                    qreg[i] = *reinterpret_cast<const frag_ab*>(
                        qp + (int64_t)(q0 + row) * qs2 + col);
                    doreg[i] = *reinterpret_cast<const frag_ab*>(
                        dop + (int64_t)(q0 + row) * gs + col);
                }
            }
        };
        auto stage_write = [&](int buf, int q0) {
#pragma unroll
            for (int i = 0; i < PPL; ++i) {
                const int p = tid + i * NT;
                if (p < NP) {
                    int row, col;
                    pkt_rc<D>(p, row, col);
                    *reinterpret_cast<frag_ab*>(
                        &qs_lds[buf][sub_idx<D>(row, col)]) = qreg[i];
                    *reinterpret_cast<frag_ab*>(
                        &dos_lds[buf][sub_idx<D>(row, col)]) = doreg[i];
                }
            }
            for (int i = tid; i < QT; i += NT) {
                lse_lds[buf][i] = lse[bh_off + q0 + i] * LOG2E;
                dlt_lds[buf][i] = delta[bh_off + q0 + i];
            }
        };

        // dk-only at D=128 sits at the register cap: staging loads move to
        // the write point so their registers don't live across the compute
        constexpr bool EARLY = !(DO_DK && !DO_DV && D == 128);
        const int n_qt = (S - q_first) / QT;
        if (n_qt > 0) {
            stage_load(q_first);
            stage_write(0, q_first);
        }
        __syncthreads();

        for (int it = 0; it < n_qt; ++it) {
            const int q0 = q_first + it * QT;
            const int cur = it & 1;
            const bool last = (it + 1 == n_qt);
            if (EARLY && !last) stage_load(q0 + QT);

            // wave-skip: a q-tile entirely above this wave's diagonal
            const bool active = full_vis || (q0 + QT - 1 >= kv0w + kv_off);
            if (active) {
                auto qs_body = [&](int qs2c) {
                    f32x16 st, dp;
#pragma unroll
                    for (int r = 0; r < 16; ++r) { st[r] = 0.f; dp[r] = 0.f; }
                    __builtin_amdgcn_s_setprio(1);
#pragma unroll
                    for (int c = 0; c < DCH; ++c) {
                        // St = Q·K^T: A = Q rows (row-major read), B = K regs
                        frag_ab aQ = *reinterpret_cast<const frag_ab*>(
                            &qs_lds[cur][sub_idx<D>(qs2c * 32 + l31,
                                                    c * 16 + 8 * hi)]);
                        st = MFMA_32x32x16(aQ, bK[c], st);
                        if (DO_DK) {
                            frag_ab aDO = *reinterpret_cast<const frag_ab*>(
                                &dos_lds[cur][sub_idx<D>(qs2c * 32 + l31,
                                                         c * 16 + 8 * hi)]);
                            dp = MFMA_32x32x16(aDO, bV[c], dp);  // dO·V^T
                        }
                    }
                    __builtin_amdgcn_s_setprio(0);
                    const int jk = n_glob + kv_off;
#pragma unroll
                    for (int r = 0; r < 16; ++r) {
                        const int q_loc = qs2c * 32 + (r & 3) + 8 * (r >> 2) +
                                          4 * hi;
                        const int iq = q0 + q_loc;
                        const float z2 = st[r] * scale2 +
                                         slope2 * (float)(jk - iq);
                        const float p = (jk <= iq)
                            ? exp2f(z2 - lse_lds[cur][q_loc]) : 0.f;
                        st[r] = p;                                   // P
                        if (DO_DK)
                            dp[r] = p * (dp[r] - dlt_lds[cur][q_loc]) * scale;
                    }
                    // pack P and dS into B-fragments over the q axis
                    frag_ab pfrag[2], dsfrag[2];
#pragma unroll
                    for (int hf = 0; hf < 2; ++hf) {
                        union { __bf16 h2[2]; uint32_t u; } wa, wb2, wc, wd;
                        if (!DO_DV) {
                            wa.h2[0] = (__bf16)dp[8 * hf + 0];
                            wa.h2[1] = (__bf16)dp[8 * hf + 1];
                            wc.h2[0] = (__bf16)dp[8 * hf + 2];
                            wc.h2[1] = (__bf16)dp[8 * hf + 3];
                            wb2.h2[0] = (__bf16)dp[8 * hf + 4];
                            wb2.h2[1] = (__bf16)dp[8 * hf + 5];
                            wd.h2[0] = (__bf16)dp[8 * hf + 6];
                            wd.h2[1] = (__bf16)dp[8 * hf + 7];
                            auto t1 = __builtin_amdgcn_permlane32_swap(
                                wa.u, wb2.u, false, false);
                            auto t2 = __builtin_amdgcn_permlane32_swap(
                                wc.u, wd.u, false, false);
                            union { frag_ab f; uint32_t u[4]; } o2;
                            o2.u[0] = t1[0];
                            o2.u[1] = t2[0];
                            o2.u[2] = t1[1];
                            o2.u[3] = t2[1];
                            dsfrag[hf] = o2.f;
                            continue;
                        }
                        wa.h2[0] = (__bf16)st[8 * hf + 0];
                        wa.h2[1] = (__bf16)st[8 * hf + 1];
                        wc.h2[0] = (__bf16)st[8 * hf + 2];
                        wc.h2[1] = (__bf16)st[8 * hf + 3];
                        wb2.h2[0] = (__bf16)st[8 * hf + 4];
                        wb2.h2[1] = (__bf16)st[8 * hf + 5];
                        wd.h2[0] = (__bf16)st[8 * hf + 6];
                        wd.h2[1] = (__bf16)st[8 * hf + 7];
                        auto s1 = __builtin_amdgcn_permlane32_swap(
                            wa.u, wb2.u, false, false);
                        auto s2 = __builtin_amdgcn_permlane32_swap(
                            wc.u, wd.u, false, false);
                        union { frag_ab f; uint32_t u[4]; } out;
                        out.u[0] = s1[0];
                        out.u[1] = s2[0];
                        out.u[2] = s1[1];
                        out.u[3] = s2[1];
                        pfrag[hf] = out.f;
                        if (!DO_DK) continue;

                        wa.h2[0] = (__bf16)dp[8 * hf + 0];
                        wa.h2[1] = (__bf16)dp[8 * hf + 1];
                        wc.h2[0] = (__bf16)dp[8 * hf + 2];
                        wc.h2[1] = (__bf16)dp[8 * hf + 3];
                        wb2.h2[0] = (__bf16)dp[8 * hf + 4];
                        wb2.h2[1] = (__bf16)dp[8 * hf + 5];
                        wd.h2[0] = (__bf16)dp[8 * hf + 6];
                        wd.h2[1] = (__bf16)dp[8 * hf + 7];
                        s1 = __builtin_amdgcn_permlane32_swap(
                            wa.u, wb2.u, false, false);
                        s2 = __builtin_amdgcn_permlane32_swap(
                            wc.u, wd.u, false, false);
                        out.u[0] = s1[0];
                        out.u[1] = s2[0];
                        out.u[2] = s1[1];
                        out.u[3] = s2[1];
                        dsfrag[hf] = out.f;
                    }
                    // dV^T += dO^T·P  and  dK^T += Q^T·dS (transpose reads)
#pragma unroll
                    for (int dsb = 0; dsb < DSUB; ++dsb) {
                        __builtin_amdgcn_s_setprio(1);
#pragma unroll
                        for (int ksq = 0; ksq < 2; ++ksq) {
                            const int g0 = qs2c * 8 + 4 * ksq + 2 * hi;
                            const int dblk = 2 * dsb + (lq & 1);
                            if (DO_DV) {
                                union { frag_ab f; bf16x4 h[2]; } adot;
                                adot.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                                    (lds_bf16x4_p)&dos_lds[cur][
                                        sub_tile_base<D>(g0, dblk) + (lane & 15) * 4]);
                                adot.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                                    (lds_bf16x4_p)&dos_lds[cur][
                                        sub_tile_base<D>(g0 + 1, dblk) + (lane & 15) * 4]);
                                accDV[DO_DV ? dsb : 0] = MFMA_32x32x16(
                                    adot.f, pfrag[ksq], accDV[DO_DV ? dsb : 0]);
                            }
                            if (DO_DK && dsb >= DK0 && dsb < DK0 + NDSB) {
                                const int dblk_k = 2 * dsb + (lq & 1);
                                union { frag_ab f; bf16x4 h[2]; } aqt;
                                aqt.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                                    (lds_bf16x4_p)&qs_lds[cur][
                                        sub_tile_base<D>(g0, dblk_k) + (lane & 15) * 4]);
                                aqt.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                                    (lds_bf16x4_p)&qs_lds[cur][
                                        sub_tile_base<D>(g0 + 1, dblk_k) + (lane & 15) * 4]);
                                const int ai = DO_DK ? dsb - DK0 : 0;
                                accDK[ai] = MFMA_32x32x16(
                                    aqt.f, dsfrag[ksq], accDK[ai]);
                            }
                        }
                        __builtin_amdgcn_s_setprio(0);
                    }
                };
                if (D == 128) {
#pragma unroll 1
                    for (int qs2c = 0; qs2c < 2; ++qs2c) qs_body(qs2c);
                } else {
                    qs_body(0);
                    qs_body(1);
                }
            }
            if (!last) {
                if (!EARLY) stage_load(q0 + QT);
                stage_write(cur ^ 1, q0 + QT);
            }
            __syncthreads();
        }
        __syncthreads();  // gq rotation reuses the LDS buffers
    }

    // epilogue: lane owns kv row n_glob; reg r is d = 8*(r>>2)+(r&3)+4*hi
    bf16* dkrow = dk + b * wb + h * wh + (int64_t)n_glob * ws;
    bf16* dvrow = dv + b * wb + h * wh + (int64_t)n_glob * ws;
#pragma unroll
    for (int dsb = 0; dsb < DSUB; ++dsb) {
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            union { __bf16 h4[4]; uint2 u; } wk2, wv2;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                if (DO_DK && dsb >= DK0 && dsb < DK0 + NDSB)
                    wk2.h4[r] = (__bf16)accDK[dsb - DK0][4 * g + r];
                if (DO_DV) wv2.h4[r] = (__bf16)accDV[dsb][4 * g + r];
            }
            if (DO_DK && dsb >= DK0 && dsb < DK0 + NDSB)
                *reinterpret_cast<uint2*>(dkrow + dsb * 32 + 8 * g + 4 * hi) = wk2.u;
            if (DO_DV)
                *reinterpret_cast<uint2*>(dvrow + dsb * 32 + 8 * g + 4 * hi) = wv2.u;
        }
    }
}

}  // namespace

void attn_bwd_into(torch::Tensor dout, torch::Tensor q,
                   torch::Tensor k, torch::Tensor v,
                   torch::Tensor o, torch::Tensor lse,
                   torch::Tensor slopes, double scale,
                   torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
                   int64_t kv_off) {
    TORCH_CHECK(dout.is_cuda() && dout.stride(3) == 1 && q.stride(3) == 1 &&
                k.stride(3) == 1 && v.stride(3) == 1 && o.stride(3) == 1,
                "attn_bwd: last dim must be contiguous");
    TORCH_CHECK(dq.stride(3) == 1 && dk.stride(3) == 1 && dv.stride(3) == 1);
    // dk and dv must share a stride layout (one stride set feeds both writes)
    TORCH_CHECK(dk.strides() == dv.strides());
    const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
    const int Hkv = k.size(1);
    TORCH_CHECK(H % Hkv == 0);
    const int kv_group = H / Hkv;
    auto delta = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));

    const int64_t rows = (int64_t)B * H * S;
    auto stream = at::cuda::getCurrentCUDAStream();

    // v2 path (32x32 MFMA, lane-local stats, tr16 transposed reads)
    static const int use_bwd_v2 = [] {
        const char* e = getenv("PG_ATTN_BWD_V2");
        return e ? atoi(e) : 1;
    }();
    // measured (r2h A/B): v2 wins at both head dims once the sub-tile loops
    // stopped unrolling (D=64: 1264 -> 1110 us; D=128: 5448 -> 5153 us on
    // the bench shapes); PG_ATTN_BWD_V2=0 forces the v1 kernels
    if (use_bwd_v2 && S % 128 == 0 && (D == 64 || D == 128)) {
#define STRV(t) t.stride(0), t.stride(1), t.stride(2)
#define BWD2_ARGS(W_OUT)                                                      \
            reinterpret_cast<const bf16*>(dout.data_ptr()),                   \
            reinterpret_cast<const bf16*>(q.data_ptr()),                      \
            reinterpret_cast<const bf16*>(k.data_ptr()),                      \
            reinterpret_cast<const bf16*>(v.data_ptr()),                      \
            lse.data_ptr<float>(), delta.data_ptr<float>(),                   \
            slopes.data_ptr<float>(), (float)scale, W_OUT,                    \
            B, H, S, (int)kv_off, kv_group,                                   \
            STRV(dout), STRV(q), STRV(k), STRV(v), STRV(dq)
#define LAUNCH_DELTA(DV)                                                      \
        hipLaunchKernelGGL((attn_bwd_delta_kernel<DV>),                       \
            dim3((rows + 3) / 4), dim3(256), 0, stream,                       \
            reinterpret_cast<const bf16*>(dout.data_ptr()),                   \
            reinterpret_cast<const bf16*>(o.data_ptr()),                      \
            delta.data_ptr<float>(), rows, H, S, STRV(dout), STRV(o))
#define LAUNCH_DQ2(DV, WV)                                                    \
    do {                                                                      \
        dim3 grid_q(S / (32 * WV), H, B);                                     \
        hipLaunchKernelGGL((attn_bwd_dq_v2_kernel<DV, WV>), grid_q,           \
            dim3(WV * WAVE_SIZE), 0, stream,                                  \
            BWD2_ARGS(reinterpret_cast<bf16*>(dq.data_ptr())));               \
    } while (0)
#define LAUNCH_BWD2(DV, WV, DKV, DVV, DKP)                                         \
    do {                                                                      \
        dim3 grid_kv(S / (32 * WV), Hkv, B);                                  \
        hipLaunchKernelGGL((attn_bwd_dkdv_v2_kernel<DV, WV, DKV, DVV, DKP>), \
            grid_kv,                                                          \
            dim3(WV * WAVE_SIZE), 0, stream,                                  \
            reinterpret_cast<const bf16*>(dout.data_ptr()),                   \
            reinterpret_cast<const bf16*>(q.data_ptr()),                      \
            reinterpret_cast<const bf16*>(k.data_ptr()),                      \
            reinterpret_cast<const bf16*>(v.data_ptr()),                      \
            lse.data_ptr<float>(), delta.data_ptr<float>(),                   \
            slopes.data_ptr<float>(), (float)scale,                           \
            reinterpret_cast<bf16*>(dk.data_ptr()),                           \
            reinterpret_cast<bf16*>(dv.data_ptr()),                           \
            B, H, S, (int)kv_off, kv_group,                                   \
            STRV(dout), STRV(q), STRV(k), STRV(v), STRV(dk));                 \
    } while (0)
        // D=64 fits the fused dk+dv kernel in 2 waves/SIMD; D=128 would
        // spill, so it runs as a dv-only and a dk-only pass (St recomputed,
        // +25% MFMA but no scratch traffic)
        static const int force_w4b = [] {
            const char* e = getenv("PG_ATTN_W4");
            return e ? atoi(e) : 0;
        }();
        if (D == 64) {
            LAUNCH_DELTA(64);
            if (S % 256 == 0 && !force_w4b) {
                LAUNCH_DQ2(64, 8);
                LAUNCH_BWD2(64, 8, true, true, -1);
            } else {
                LAUNCH_DQ2(64, 4);
                LAUNCH_BWD2(64, 4, true, true, -1);
            }
        } else {
            // fused dk+dv spills at D=128 (256+144 regs wanted); run as a
            // dv-only pass plus two half-width dk passes, all 8-wave
            LAUNCH_DELTA(128);
            if (S % 256 == 0) {
                LAUNCH_DQ2(128, 8);
                LAUNCH_BWD2(128, 8, false, true, -1);
                LAUNCH_BWD2(128, 8, true, false, 0);
                LAUNCH_BWD2(128, 8, true, false, 1);
            } else {
                LAUNCH_DQ2(128, 4);
                LAUNCH_BWD2(128, 4, false, true, -1);
                LAUNCH_BWD2(128, 4, true, false, 0);
                LAUNCH_BWD2(128, 4, true, false, 1);
            }
        }
#undef LAUNCH_BWD2
#undef LAUNCH_DQ2
#undef LAUNCH_DELTA
#undef BWD2_ARGS
#undef STRV
        HIP_CHECK_LAUNCH();
        return;
    }

    const int dkdv_waves = (S % 128 == 0) ? 8 : 4;
    dim3 grid_a(S / (16 * dkdv_waves), Hkv, B);
    dim3 grid_b(S / (16 * dkdv_waves), H, B);

#define STR3(t) t.stride(0), t.stride(1), t.stride(2)
#define BWD_DKDV_ARGS                                                         \
            reinterpret_cast<const bf16*>(dout.data_ptr()),                   \
            reinterpret_cast<const bf16*>(q.data_ptr()),                      \
            reinterpret_cast<const bf16*>(k.data_ptr()),                      \
            reinterpret_cast<const bf16*>(v.data_ptr()),                      \
            lse.data_ptr<float>(), delta.data_ptr<float>(),                   \
            slopes.data_ptr<float>(), (float)scale,                           \
            reinterpret_cast<bf16*>(dk.data_ptr()),                           \
            reinterpret_cast<bf16*>(dv.data_ptr()), B, H, S, (int)kv_off,     \
            kv_group,                                                         \
            STR3(dout), STR3(q), STR3(k), STR3(v), STR3(dk)
#define BWD_DQ_ARGS                                                           \
            reinterpret_cast<const bf16*>(dout.data_ptr()),                   \
            reinterpret_cast<const bf16*>(q.data_ptr()),                      \
            reinterpret_cast<const bf16*>(k.data_ptr()),                      \
            reinterpret_cast<const bf16*>(v.data_ptr()),                      \
            lse.data_ptr<float>(), delta.data_ptr<float>(),                   \
            slopes.data_ptr<float>(), (float)scale,                           \
            reinterpret_cast<bf16*>(dq.data_ptr()), B, H, S, (int)kv_off,     \
            kv_group,                                                         \
            STR3(dout), STR3(q), STR3(k), STR3(v), STR3(dq)
#define LAUNCH_BWD(DV)                                                        \
    do {                                                                      \
        hipLaunchKernelGGL((attn_bwd_delta_kernel<DV>),                       \
            dim3((rows + 3) / 4), dim3(256), 0, stream,                       \
            reinterpret_cast<const bf16*>(dout.data_ptr()),                   \
            reinterpret_cast<const bf16*>(o.data_ptr()),                      \
            delta.data_ptr<float>(), rows, H, S, STR3(dout), STR3(o));        \
        if (dkdv_waves == 8) {                                                \
            hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DV, 32, 8>),             \
                grid_a, dim3(512), 0, stream, BWD_DKDV_ARGS);                 \
        } else {                                                              \
            hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DV, 32, 4>),             \
                grid_a, dim3(256), 0, stream, BWD_DKDV_ARGS);                 \
        }                                                                     \
        if (dkdv_waves == 8) {                                                \
            hipLaunchKernelGGL((attn_bwd_dq_kernel<DV, 32, 8>),               \
                grid_b, dim3(512), 0, stream, BWD_DQ_ARGS);                   \
        } else {                                                              \
            hipLaunchKernelGGL((attn_bwd_dq_kernel<DV, 32, 4>),               \
                grid_b, dim3(256), 0, stream, BWD_DQ_ARGS);                   \
        }                                                                     \
    } while (0)

    if (D == 64) {
        LAUNCH_BWD(64);
    } else {
        TORCH_CHECK(D == 128, "attn_bwd: head dim 64 or 128");
        LAUNCH_BWD(128);
    }
#undef LAUNCH_BWD
#undef BWD_DKDV_ARGS
#undef BWD_DQ_ARGS
#undef STR3
    HIP_CHECK_LAUNCH();
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    torch::Tensor slopes, double scale,
                                    int64_t kv_off) {
    const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
    const int Hkv = k.size(1);
    // grads physically [B, S, H, D] (the layout the fused-qkv backward wants)
    auto dq_phys = torch::empty({B, S, H, D}, q.options());
    auto dk_phys = torch::empty({B, S, Hkv, D}, q.options());
    auto dv_phys = torch::empty({B, S, Hkv, D}, q.options());
    auto dq = dq_phys.permute({0, 2, 1, 3});
    auto dk = dk_phys.permute({0, 2, 1, 3});
    auto dv = dv_phys.permute({0, 2, 1, 3});
    attn_bwd_into(dout, q, k, v, o, lse, slopes, scale, dq, dk, dv, kv_off);
    return {dq, dk, dv};
}
