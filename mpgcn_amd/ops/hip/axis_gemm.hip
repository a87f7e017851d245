// axis_gemm — the 2-D graph-convolution contraction engine (gfx950 MFMA).
//
// Computes, per instance (batched):   OUT[m, q] = sum_k AT[m, k] * X[k, q]
// with flexible (grouped-stride) addressing on X rows/cols and OUT cols, so one
// kernel serves every contraction of the factored BDGCN layer (see
// mpgcn_amd/ops/eager.py for the algebra; reference call sites MPGCN.py:28-50):
//
//   mode-1 fwd : U[b,m,d,o,l] = sum_n  Go^T[m,n]      X[b,n,d,l]       (K1)
//   mode-2 fwd : Y[b,m,d,h]   = sum_cs A2T[d,cs]      V[b,m,cs,h]      (K2, +bias+ReLU)
//   bwd dV     : dV[b,m,cs,h] = sum_d  A2[cs,d]       dY[b,m,d,h]
//   bwd dX     : dX[b,n,d,l]  = sum_om A3T[n,om]      dU[b,m,d,o,l]
//
// The graph operand AT is always passed (M, K) row-major (k contiguous) — the
// Python layer pre-permutes the tiny graph tensors — so the A-tile stages into
// LDS as a straight vectorized copy; the X-tile is transposed into an
// [n][k]-major LDS image during staging so MFMA fragments on both operands are
// contiguous ds_read_b128 (bf16) reads. LDS rows are padded by 16 B: the
// resulting 144 B row stride makes 16-lane b128 fragment reads bank-conflict-free
// (row*36 mod 64 covers all 64 banks).
#include "common.hpp"
#include "params.hpp"

__device__ __forceinline__ long xrow_off(const AxisGemmParams& p, int k) {
    return p.kdiv > 1 ? (long)(k / p.kdiv) * p.k_hi + (long)(k % p.kdiv) * p.k_lo
                      : (long)k * p.k_lo;
}
__device__ __forceinline__ long xcol_off(const AxisGemmParams& p, int q) {
    return p.qdiv ? (long)(q / p.qdiv) * p.q_hi + (long)(q % p.qdiv) : (long)q;
}
__device__ __forceinline__ long ocol_off(const AxisGemmParams& p, int q) {
    return p.ogdiv ? (long)(q / p.ogdiv) * p.og_hi + (long)(q % p.ogdiv) : (long)q;
}

template <typename T, int BM, int BN, int BK, int WVM, int WVN>
__launch_bounds__(256) __global__ void axis_gemm_kernel(AxisGemmParams p) {
    using MT = MfmaTraits<T>;
    constexpr int CH = 16 / sizeof(T);  // elements per 16-byte chunk
    constexpr int PAD = MT::LDS_PAD;
    constexpr int WM = BM / WVM, WN = BN / WVN;
    constexpr int AM = WM / 16, AN = WN / 16;
    static_assert(WVM * WVN == 4, "4 waves per block");

    __shared__ T ldsA[BM][BK + PAD];
    __shared__ T ldsB[BN][BK + PAD];

    const T* __restrict__ A = (const T*)p.AT;
    const T* __restrict__ X = (const T*)p.X;
    T* __restrict__ O = (T*)p.OUT;

    const int inst = blockIdx.y;
    const long a_base = (long)(inst / p.a_div) * p.a_bs1 + (long)(inst % p.a_div) * p.a_bs2;
    const long x_base = (long)(inst / p.x_div) * p.x_bs1 + (long)(inst % p.x_div) * p.x_bs2;
    const long o_base = (long)(inst / p.o_div) * p.o_bs1 + (long)(inst % p.o_div) * p.o_bs2;

    const int tm = blockIdx.x / p.tiles_l;
    const int tl = blockIdx.x % p.tiles_l;
    const int m0 = tm * BM;
    const int l0 = tl * BN;

    const int tid = threadIdx.x;
    const int w = tid / WAVE, lane = tid % WAVE;
    const int wm = (w / WVN) * WM, wn = (w % WVN) * WN;
    const int lrow = lane & 15, kgrp = lane >> 4;

    f32x4 acc[AM][AN] = {};

    for (int kt = 0; kt < p.K; kt += BK) {
        // ---- stage A tile (straight copy, k contiguous) ----
        constexpr int A_CPR = BK / CH;  // chunks per row
        for (int idx = tid; idx < BM * A_CPR; idx += 256) {
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
            *(Chunk16*)&ldsA[row][cc * CH] = val;
        }
        // ---- stage X tile, transposed to [n][k] ----
        constexpr int B_CPR = BN / CH;
        for (int idx = tid; idx < BK * B_CPR; idx += 256) {
            const int krow = idx / B_CPR, qc = idx % B_CPR;
            const int k = kt + krow;
            const int q0 = l0 + qc * CH;
            alignas(16) T tmp[CH];
            for (int i = 0; i < CH; ++i) tmp[i] = (T)0.f;
            if (k < p.K) {
                const long rbase = x_base + xrow_off(p, k);
                if (p.x_vec && q0 + CH <= p.L) {
                    Chunk16 c = *(const Chunk16*)(X + rbase + xcol_off(p, q0));
                    *(Chunk16*)tmp = c;
                } else {
                    for (int i = 0; i < CH; ++i)
                        if (q0 + i < p.L) tmp[i] = X[rbase + xcol_off(p, q0 + i)];
                }
            }
            // lane-rotated write order de-conflicts the transpose (see
            // red_gemm.hip): stride between lanes' rows is ~0 mod 32 banks,
            // rotating the element index spreads each instruction's writes
#pragma unroll
            for (int i = 0; i < CH; ++i) {
                const int j = (i + tid) % CH;
                ldsB[qc * CH + j][krow] = tmp[j];
            }
        }
        __syncthreads();

        // ---- MFMA over the tile ----
        for (int kk = 0; kk < BK; kk += MT::MFMA_K) {
            typename MT::frag_t af[AM], bf[AN];
#pragma unroll
            for (int mf = 0; mf < AM; ++mf)
                af[mf] = *(const typename MT::frag_t*)
                    &ldsA[wm + mf * 16 + lrow][kk + kgrp * MT::FRAG_ELEMS];
#pragma unroll
            for (int nf = 0; nf < AN; ++nf)
                bf[nf] = *(const typename MT::frag_t*)
                    &ldsB[wn + nf * 16 + lrow][kk + kgrp * MT::FRAG_ELEMS];
#pragma unroll
            for (int mf = 0; mf < AM; ++mf)
#pragma unroll
                for (int nf = 0; nf < AN; ++nf)
                    acc[mf][nf] = MT::mfma(af[mf], bf[nf], acc[mf][nf]);
        }
        __syncthreads();
    }

    // ---- epilogue: bias + activation + strided store ----
#pragma unroll
    for (int mf = 0; mf < AM; ++mf) {
#pragma unroll
        for (int nf = 0; nf < AN; ++nf) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + mf * 16 + kgrp * 4 + r;
                const int q = l0 + wn + nf * 16 + lrow;
                if (m < p.M && q < p.L) {
                    float v = acc[mf][nf][r];
                    if (p.bias) v += p.bias[p.bias_mod ? q % p.bias_mod : q];
                    if (p.relu) v = fmaxf(v, 0.f);
                    O[o_base + (long)m * p.o_row + ocol_off(p, q)] = from_f32<T>(v);
                }
            }
        }
    }
}

extern "C" void axis_gemm_launch(AxisGemmParams p, int instances, int is_f32,
                                 hipStream_t stream) {
    constexpr int BM = 128, BK = 64;
    const int BN = (p.L >= 48) ? 64 : 32;
    const int tiles_m = (p.M + BM - 1) / BM;
    p.tiles_l = (p.L + BN - 1) / BN;
    dim3 grid(tiles_m * p.tiles_l, instances), block(256);
    if (!is_f32) {
        if (BN == 64)
            axis_gemm_kernel<__bf16, BM, 64, BK, 2, 2><<<grid, block, 0, stream>>>(p);
        else
            axis_gemm_kernel<__bf16, BM, 32, BK, 4, 1><<<grid, block, 0, stream>>>(p);
    } else {
        if (BN == 64)
            axis_gemm_kernel<float, BM, 64, BK, 2, 2><<<grid, block, 0, stream>>>(p);
        else
            axis_gemm_kernel<float, BM, 32, BK, 4, 1><<<grid, block, 0, stream>>>(p);
    }
}
