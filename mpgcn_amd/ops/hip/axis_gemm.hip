// axis_gemm — the 2-D graph-convolution contraction engine (gfx950 MFMA).
//
// Computes, per instance (batched):   OUT[m, q] = sum_k AT[m, k] * X[k, q]
// with flexible (grouped-stride) addressing on X rows/cols and OUT cols, so one
// kernel serves every contraction of the factored BDGCN layer (see
// mpgcn_amd/ops/eager.py for the algebra; reference call sites MPGCN.py:28-50):
//
//   mode-1 fwd : U[b,m,d,o,l]  = sum_n  Go^T[m,n]   X[b,n,d,l]       (K1)
//   mode-2 fwd : Y[b,m,d,h]    = sum_cs A2T[d,cs]   V[b,m,cs,h]      (K2, +bias+ReLU)
//   bwd dV     : dV[b,m,cs,h]  = sum_d  A2[cs,d]    dY[b,m,d,h]
//   bwd dX     : dX[b,n,d,l]   = sum_om A3T[n,om]   dU[b,m,d,o,l]
//
// Structure: LDS-tiled K loop. bf16 instantiations run SINGLE-buffered —
// these contractions are global-latency-bound (PMC: 67% WAIT_ANY), so the
// halved LDS footprint (2x resident blocks/CU) beats the double-buffered
// 2-phase pipeline (A/B-measured; profiles/SUMMARY.md). The f32 test-oracle
// path keeps the classic double-buffered prefetch schedule.
// The graph operand AT is (M, K) row-major (the
// Python layer pre-permutes the tiny graph tensors), so the A-tile stages as a
// straight vectorized copy; the X-tile transposes into an [n][k]-major image
// with lane-rotated element order (spreads write banks over 4*{0..7}). LDS
// rows are padded 16 B so 16-lane b128 fragment reads are conflict-free.
#include "common.hpp"
#include "params.hpp"

__device__ __forceinline__ long xrow_off(const AxisGemmParams& p, int k) {
    return p.k_base +
           (p.kdiv > 1 ? (long)(k / p.kdiv) * p.k_hi + (long)(k % p.kdiv) * p.k_lo
                       : (long)k * p.k_lo);
}
__device__ __forceinline__ long xcol_off(const AxisGemmParams& p, int q) {
    return p.qdiv ? (long)(q / p.qdiv) * p.q_hi + (long)(q % p.qdiv) : (long)q;
}
__device__ __forceinline__ long ocol_off(const AxisGemmParams& p, int q) {
    return p.ogdiv ? (long)(q / p.ogdiv) * p.og_hi + (long)(q % p.ogdiv) : (long)q;
}

template <typename T, int BM, int BN, int BK, int WVM, int WVN, int BUFS = 2,
          bool VEC_ONLY = false, typename OT = T, typename OT2 = OT>
__launch_bounds__(WVM * WVN * 64) __global__ void axis_gemm_kernel(AxisGemmParams p) {
    using MT = MfmaTraits<T>;
    constexpr int CH = 16 / sizeof(T);
    constexpr int PAD = MT::LDS_PAD;
    constexpr int WM = BM / WVM, WN = BN / WVN;
    constexpr int AM = WM / 16, AN = WN / 16;
    constexpr int NT = WVM * WVN * 64;  // block threads (4 or 8 waves)
    static_assert(WVM * WVN == 4 || WVM * WVN == 8);

    __shared__ T ldsA[BUFS][BM][BK + PAD];
    __shared__ T ldsB[BUFS][BN][BK + PAD];

    const T* __restrict__ A = (const T*)p.AT;
    const T* __restrict__ X = (const T*)p.X;
    OT* __restrict__ O = (OT*)p.OUT;
    OT2* __restrict__ O2 = (OT2*)p.OUT2;  // optional second copy (fp8 twins)

    const int inst = blockIdx.y;
    const long a_base = (long)(inst / p.a_div) * p.a_bs1 + (long)(inst % p.a_div) * p.a_bs2;
    const long x_base = (long)(inst / p.x_div) * p.x_bs1 + (long)(inst % p.x_div) * p.x_bs2;
    const long o_base = (long)(inst / p.o_div) * p.o_bs1 + (long)(inst % p.o_div) * p.o_bs2;
    const T* __restrict__ CS = (const T*)p.CSUB;
    const long cs_base = p.CSUB
        ? (long)(inst / p.cs_div) * p.cs_bs1 + (long)(inst % p.cs_div) * p.cs_bs2
        : 0;

    const int tm = blockIdx.x / p.tiles_l;
    const int tl = blockIdx.x % p.tiles_l;
    const int m0 = tm * BM;
    const int l0 = tl * BN;

    const int tid = threadIdx.x;
    const int w = tid / WAVE, lane = tid % WAVE;
    const int wm = (w / WVN) * WM, wn = (w % WVN) * WN;
    const int lrow = lane & 15, kgrp = lane >> 4;

    f32x4 acc[AM][AN] = {};

    auto stage = [&](int buf, int kt) {
        // ---- A tile: straight vectorized copy (k contiguous) ----
        constexpr int A_CPR = BK / CH;
        for (int idx = tid; idx < BM * A_CPR; idx += NT) {
            const int row = idx / A_CPR, cc = idx % A_CPR;
            const int m = m0 + row, k = kt + cc * CH;
            Chunk16 val = {};
            if (m < p.M) {
                const T* src = A + a_base + (long)m * p.K + k;
                if (p.a_vec && k + CH <= p.K) {
                    val = *(const Chunk16*)src;
                } else {
                    T* d = (T*)&val;
                    for (int i = 0; i < CH; ++i) d[i] = (k + i < p.K) ? src[i] : (T)0.f;
                }
            }
            *(Chunk16*)&ldsA[buf][row][cc * CH] = val;
        }
        // ---- X tile, transposed to [n][k], lane-rotated write order ----
        // 256 % B_CPR == 0, so each thread's column chunk qc is CONSTANT across
        // its grid-stride chunks: the q-side division in xcol_off hoists out of
        // the loop entirely (it otherwise dominates issue time as a magic-
        // number division sequence per chunk per K-tile).
        constexpr int B_CPR = BN / CH;
        static_assert(NT % B_CPR == 0);
        const int qc = tid % B_CPR;
        const int q0 = l0 + qc * CH;
        const bool qvec = p.x_vec && q0 + CH <= p.L;
        const long xq = qvec ? xcol_off(p, q0) : 0;
        for (int idx = tid; idx < BK * B_CPR; idx += NT) {
            const int krow = idx / B_CPR;
            const int k = kt + krow;
            alignas(16) T tmp[CH];
            for (int i = 0; i < CH; ++i) tmp[i] = (T)0.f;
            if (k < p.K) {
                const long rbase = x_base + xrow_off(p, k);
                if (VEC_ONLY || qvec) {
                    // VEC_ONLY instantiations (fp8 probe) compile the vector
                    // path alone: the CH-wide scalar fallback's address set
                    // spills at CH = 16 bytes/lane
                    *(Chunk16*)tmp = *(const Chunk16*)(X + rbase + xq);
                } else if constexpr (!VEC_ONLY) {
                    for (int i = 0; i < CH; ++i)
                        if (q0 + i < p.L) tmp[i] = X[rbase + xcol_off(p, q0 + i)];
                }
            }
#pragma unroll
            for (int i = 0; i < CH; ++i) {
                const int j = (i + tid) % CH;
                ldsB[buf][qc * CH + j][krow] = tmp[j];
            }
        }
    };

    const int ktiles = (p.K + BK - 1) / BK;
    int cur = 0;
    if (BUFS == 2) {
        stage(0, 0);
        __syncthreads();
    }
    for (int t = 0; t < ktiles; ++t) {
        if (BUFS == 2) {
            if (t + 1 < ktiles) stage(cur ^ 1, (t + 1) * BK);  // prefetch next
        } else {
            stage(0, t * BK);
            __syncthreads();
        }
#pragma unroll
        for (int kk = 0; kk < BK; kk += MT::MFMA_K) {
            typename MT::frag_t af[AM], bf[AN];
#pragma unroll
            for (int mf = 0; mf < AM; ++mf)
                af[mf] = *(const typename MT::frag_t*)
                    &ldsA[cur][wm + mf * 16 + lrow][kk + kgrp * MT::FRAG_ELEMS];
#pragma unroll
            for (int nf = 0; nf < AN; ++nf)
                bf[nf] = *(const typename MT::frag_t*)
                    &ldsB[cur][wn + nf * 16 + lrow][kk + kgrp * MT::FRAG_ELEMS];
#pragma unroll
            for (int mf = 0; mf < AM; ++mf)
#pragma unroll
                for (int nf = 0; nf < AN; ++nf)
                    acc[mf][nf] = MT::mfma(af[mf], bf[nf], acc[mf][nf]);
        }
        __syncthreads();
        if (BUFS == 2) cur ^= 1;
    }

    // ---- epilogue: v = (alpha*acc + cs_beta*csub)*scale + bias; act ----
    // per-lane q depends only on nf: hoist the column-offset division, the
    // bias load and the csub column base out of the (mf, r) loops
    const float osc = p.scale ? *p.scale : 1.f;  // fp8 gradient descale
    const float alpha = p.alpha == 0.f ? 1.f : p.alpha;
#pragma unroll
    for (int nf = 0; nf < AN; ++nf) {
        const int q = l0 + wn + nf * 16 + lrow;
        if (q >= p.L) continue;
        const long oc = o_base + ocol_off(p, q);
        const long cs_col = CS ? cs_base + xcol_off(p, q) : 0;
        const float bv = p.bias ? p.bias[p.bias_mod ? q % p.bias_mod : q] : 0.f;
#pragma unroll
        for (int mf = 0; mf < AM; ++mf) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + mf * 16 + kgrp * 4 + r;
                if (m < p.M) {
                    float v = alpha * acc[mf][nf][r];
                    if (CS)  // identity-support term / polynomial recurrence
                        v += p.cs_beta * to_f32(CS[cs_col + (long)m * p.cs_row]);
                    v = v * osc + bv;
                    if (p.relu) v = fmaxf(v, 0.f);
                    const long orow = p.o_mdiv
                        ? (long)(m / p.o_mdiv) * p.o_m_hi +
                              (long)(m % p.o_mdiv) * p.o_m_lo + p.o_m_base
                        : (long)m * p.o_row;
                    O[oc + orow] = from_f32<OT>(v);
                    if (p.OUT2) O2[oc + orow] = from_f32<OT2>(v);
                }
            }
        }
    }
}

// fp8 engine: same schedule at BK=128 — equal LDS footprint to the bf16
// BK=64 tile but half the staged bytes per K element, i.e. half the stage
// count at the same per-stage cost (measured 1.43x mode-2 / 1.30x mode-1,
// profiles/SUMMARY.md "fp8 probe"). out_kind selects the training epilogues:
//   0: fp8 out only (the round-1 probe contract)
//   1: fp8 out + bf16 twin  (mode-1: U8 feeds the fp8 projection, U_bf16 is
//      the saved backward operand — forward stays fp8, backward stays bf16)
//   2: bf16 out + fp8 twin  (mode-2: Y_bf16 is the autograd output/ReLU mask,
//      Y8 feeds the next layer's mode-1 without a separate quantize pass)
extern "C" void axis_gemm_fp8_launch(AxisGemmParams p, int instances,
                                     int out_kind, hipStream_t stream) {
    constexpr int BN = 256, BM = 256;
    const int tiles_m = (p.M + BM - 1) / BM;
    p.tiles_l = (p.L + BN - 1) / BN;
    dim3 grid(tiles_m * p.tiles_l, instances);
    if (out_kind == 1)
        axis_gemm_kernel<unsigned char, 256, 256, 128, 4, 2, 1, true,
                         unsigned char, __bf16>
            <<<grid, dim3(512), 0, stream>>>(p);
    else if (out_kind == 2)
        axis_gemm_kernel<unsigned char, 256, 256, 128, 4, 2, 1, true,
                         __bf16, unsigned char>
            <<<grid, dim3(512), 0, stream>>>(p);
    else
        axis_gemm_kernel<unsigned char, 256, 256, 128, 4, 2, 1, true>
            <<<grid, dim3(512), 0, stream>>>(p);
}

extern "C" void axis_gemm_launch(AxisGemmParams p, int instances, int is_f32,
                                 hipStream_t stream) {
    constexpr int BK = 64;
    // (A/B round 2: a 128x128 4-wave tile for short-K contractions measured
    // -6% at b32 and far worse at b128 — the byte-volume halving of the 256
    // tile beats stage-latency overlap even at 2 K-stages; null kept in
    // profiles/SUMMARY.md)
    int BN = (p.L >= 96 && p.M >= 96) ? 128 : (p.L >= 48 ? 64 : 32);
    if (!is_f32 && p.M >= 192 && p.L >= 192) BN = 256;
    if (is_f32 && BN > 64) BN = 64;  // f32 LDS budget caps the tile
    const int BM = (BN >= 256) ? 256 : 128;
    const int tiles_m = (p.M + BM - 1) / BM;
    p.tiles_l = (p.L + BN - 1) / BN;
    dim3 grid(tiles_m * p.tiles_l, instances), block(256);
    if (!is_f32) {
        // single-buffer: half the LDS -> twice the blocks/CU; these kernels
        // are global-latency-bound (PMC: 67% WAIT_ANY), so occupancy beats
        // the 2-buffer pipeline here
        if (BN == 256) {
            // 256x256 tile (8 waves): halves BOTH operands' total load bytes
            // (the binding resource is the per-CU load path, ~10 B/cyc/CU)
            axis_gemm_kernel<__bf16, 256, 256, BK, 4, 2, 1>
                <<<grid, dim3(512), 0, stream>>>(p);
        } else if (BN == 128)
            axis_gemm_kernel<__bf16, 128, 128, BK, 2, 2, 1><<<grid, block, 0, stream>>>(p);
        else if (BN == 64)
            axis_gemm_kernel<__bf16, 128, 64, BK, 2, 2, 1><<<grid, block, 0, stream>>>(p);
        else
            axis_gemm_kernel<__bf16, 128, 32, BK, 4, 1, 1><<<grid, block, 0, stream>>>(p);
    } else {
        // f32: halve BK to keep the double-buffered LDS within budget
        if (BN >= 64)
            axis_gemm_kernel<float, 128, 64, 32, 2, 2><<<grid, block, 0, stream>>>(p);
        else
            axis_gemm_kernel<float, 128, 32, 32, 4, 1><<<grid, block, 0, stream>>>(p);
    }
}
