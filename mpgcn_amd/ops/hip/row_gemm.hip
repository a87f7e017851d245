// row_gemm — fused row-streaming GEMM: OUT[R, N] = act(X[R, K] @ W[K, N] + bias).
//
// For the memory-bound "many tiny rows" GEMMs of the MPGCN stack: the BDGCN
// projection V = U_flat @ Wre (R = b*N^2 rows, K = S*C, N = S*H; reference op
// MPGCN.py:44-49), its backward dU = dV @ Wre^T, the FC head Linear+ReLU
// (MPGCN.py:74-76,107), and the LSTM backward dh_prev = dgates @ W_hh.
//
// W is staged once per block into LDS transposed ([n][k]-major) so B-fragments
// are contiguous reads; X rows feed A-fragments straight from global (each row
// is a short contiguous K-vector). Grid-stride loop over 64-row tiles.
#include "common.hpp"
#include "params.hpp"

// 4 waves * 16 rows = 64 rows per block; NF16 = ceil(N/16) column fragments.
template <typename T>
__launch_bounds__(256) __global__ void row_gemm_kernel(RowGemmParams p, int nf16) {
    using MT = MfmaTraits<T>;
    constexpr int CH = 16 / sizeof(T);
    const int KP = p.K + MT::LDS_PAD;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    T* ldsWT = (T*)smem;  // [nf16*16][K + PAD]

    const T* __restrict__ X = (const T*)p.X;
    const T* __restrict__ X2 = (const T*)p.X2;  // identity block (k < k0)
    const int k0 = p.k0, kb = p.K - p.k0;       // part widths (X2 | X)
    const T* __restrict__ W = (const T*)p.W;
    T* __restrict__ O = (T*)p.OUT;
    unsigned char* __restrict__ O8 = (unsigned char*)p.OUT8;
    const float qs = p.q_scale ? *p.q_scale : 1.f;
    float amax = 0.f;  // thread-local |v| max (fp8 delayed-scaling record)

    const int tid = threadIdx.x;
    // ---- stage W transposed: ldsWT[n][k] = W[k][n] (one time per block) ----
    for (int idx = tid; idx < p.K * p.N; idx += 256) {
        const int k = idx / p.N, n = idx % p.N;
        ldsWT[n * KP + k] = W[(long)k * p.N + n];
    }
    // zero-pad the tail fragment rows (n >= N) and k tail
    for (int idx = tid; idx < nf16 * 16 * KP; idx += 256) {
        const int n = idx / KP, k = idx % KP;
        if (n >= p.N || k >= p.K) ldsWT[n * KP + k] = (T)0.f;
    }
    __syncthreads();

    const int w = tid / WAVE, lane = tid % WAVE;
    const int lrow = lane & 15, kgrp = lane >> 4;
    const int kfrags = (p.K + MT::MFMA_K - 1) / MT::MFMA_K;

    for (long r0 = (long)blockIdx.x * 64 + w * 16; r0 < p.R;
         r0 += (long)gridDim.x * 64) {
        const long row = r0 + lrow;
        const bool row_ok = row < p.R;
        f32x4 acc[8] = {};  // up to 8 column fragments (N <= 128)

        for (int kf = 0; kf < kfrags; ++kf) {
            const int k = kf * MT::MFMA_K + kgrp * MT::FRAG_ELEMS;
            typename MT::frag_t af;
            // split-source rows: [X2 row (k0) | X row (K - k0)]. Callers
            // guarantee k0 % FRAG_ELEMS == 0 when X2 is set, so a fragment
            // never straddles the boundary on the vector path.
            const T* src = (X2 && k < k0) ? X2 : X;
            const long base = (X2 && k < k0) ? row * (long)k0 + k
                              : (X2 ? row * (long)kb + (k - k0)
                                    : row * (long)p.K + k);
            if (row_ok && k + MT::FRAG_ELEMS <= p.K && p.x_vec) {
                af = *(const typename MT::frag_t*)&src[base];
            } else {
                alignas(16) T tmp[MT::FRAG_ELEMS];
#pragma unroll
                for (int i = 0; i < MT::FRAG_ELEMS; ++i)
                    tmp[i] = (row_ok && k + i < p.K) ? src[base + i] : (T)0.f;
                af = *(const typename MT::frag_t*)tmp;
            }
            for (int nf = 0; nf < nf16; ++nf) {
                const typename MT::frag_t bf = *(const typename MT::frag_t*)
                    &ldsWT[(nf * 16 + lrow) * KP + kf * MT::MFMA_K +
                           kgrp * MT::FRAG_ELEMS];
                acc[nf] = MT::mfma(af, bf, acc[nf]);
            }
        }

        for (int nf = 0; nf < nf16; ++nf) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const long m = r0 + kgrp * 4 + r;
                const int n = nf * 16 + lrow;
                if (m < p.R && n < p.N) {
                    float v = acc[nf][r];
                    if (p.bias) v += p.bias[n];
                    if (p.relu) v = fmaxf(v, 0.f);
                    if (O) O[m * p.o_row + p.o_off + n] = from_f32<T>(v);
                    if (O8) {
                        amax = fmaxf(amax, fabsf(v));
                        O8[m * p.o_row + p.o_off + n] =
                            from_f32<unsigned char>(v * qs);
                    }
                }
            }
        }
    }

    if (p.amax_out) {
        // wave max -> one atomic per wave (f32 >= 0: uint bit order matches)
#pragma unroll
        for (int s = 32; s >= 1; s >>= 1) amax = fmaxf(amax, __shfl_xor(amax, s));
        if (lane == 0)
            atomicMax((unsigned int*)p.amax_out, __float_as_uint(amax));
    }
}

extern "C" void row_gemm_launch(RowGemmParams p, int is_f32, hipStream_t stream) {
    const int nf16 = (p.N + 15) / 16;
    const int elem = is_f32 ? 4 : 2;
    const int pad = is_f32 ? 4 : 8;
    const size_t smem = (size_t)nf16 * 16 * (p.K + pad) * elem + 64;  // +64 B slack: B-frag k-tail reads may overrun the last LDS row
    long tiles = (p.R + 63) / 64;
    if (tiles > 16384) tiles = 16384;
    dim3 grid((unsigned)tiles), block(256);
    if (!is_f32)
        row_gemm_kernel<__bf16><<<grid, block, smem, stream>>>(p, nf16);
    else
        row_gemm_kernel<float><<<grid, block, smem, stream>>>(p, nf16);
}

// fp8 e4m3 path (fp8-forward training mode): X rows and W both fp8, fp8 out.
// Same schedule; the OCP fp8 16x16x32 MFMA runs at the bf16 rate and the row
// stream is half the bytes — this GEMM is memory-bound on the X rows
// (profiles/SUMMARY.md: 2.2 TB/s bf16), so byte halving is the lever.
extern "C" void row_gemm_fp8_launch(RowGemmParams p, hipStream_t stream) {
    const int nf16 = (p.N + 15) / 16;
    const size_t smem = (size_t)nf16 * 16 * (p.K + 16) * 1 + 64;
    long tiles = (p.R + 63) / 64;
    if (tiles > 16384) tiles = 16384;
    dim3 grid((unsigned)tiles), block(256);
    row_gemm_kernel<unsigned char><<<grid, block, smem, stream>>>(p, nf16);
}
