// red_gemm — fused tall-skinny reduction GEMM for weight gradients (gfx950).
//
//   out[k, n]  = sum_r X[r, k] * Y[r, n]        (f32, atomically accumulated)
//   colsum[k]  = sum_r X[r, k]                  (optional; bias gradients)
//   xdot[k]    = sum_r X[r, k] * xv[r]          (optional; LSTM w_ih gradient)
//
// Replaces the rocBLAS calls the backward pass would otherwise make for
//   dWre^T = dV^T @ U          (BDGCN projection-weight grad, R = B*N^2 rows)
//   dW_hh  = dgates^T @ h_prev (+ dbias = colsum, dw_ih = xdot) per LSTM step
// — profiling showed rocBLAS runs these 2M-row reductions at ~4% of HBM
// bandwidth (65% of total step time); this kernel streams X and Y exactly once.
//
// Both operands are staged transposed into LDS ([col][row] images) so the MFMA
// A- and B-fragments are contiguous b128 reads; each block grid-strides over
// 64-row chunks and atomically adds its f32 partials once at the end.
#include "common.hpp"
#include "params.hpp"

typedef __attribute__((ext_vector_type(2))) float f32x2;

// YF8: the Y operand is e4m3 fp8 in global memory (the fp8-forward mode's
// U8 tensor — saving it fp8-only halves both the mode-1 write traffic and
// this kernel's Y read bytes); converted to bf16 during LDS staging, MFMA
// unchanged. v_cvt_pk_f32_fp8 unpacks byte pairs.
template <typename T, int AKF, int ANF, bool YF8 = false>
__launch_bounds__(256) __global__ void red_gemm_kernel(RedGemmParams p) {
    using MT = MfmaTraits<T>;
    constexpr int CH = 16 / sizeof(T);
    // rows per chunk: smaller -> less LDS -> more resident blocks (the
    // kernel is latency-bound; occupancy is the lever)
    constexpr int RCH = 64;
    constexpr int PAD = MT::LDS_PAD;
    constexpr int KMAX = AKF * 32, NMAX = ANF * 32;

    __shared__ T ldsXT[KMAX][RCH + PAD];
    __shared__ T ldsYT[NMAX][RCH + PAD];
    __shared__ T ldsXV[RCH];

    const T* __restrict__ X = (const T*)p.X;
    const T* __restrict__ Y = (const T*)p.Y;
    const T* __restrict__ XV = (const T*)p.xvec;

    const int tid = threadIdx.x;
    const int w = tid / WAVE, lane = tid % WAVE;
    const int wk = w / 2, wn = w % 2;
    const int lrow = lane & 15, kgrp = lane >> 4;

    f32x4 acc[AKF][ANF] = {};
    float cs = 0.f, xd = 0.f;

    for (long r0 = (long)blockIdx.x * RCH; r0 < p.R; r0 += (long)gridDim.x * RCH) {
        // stage X chunk transposed: ldsXT[k][r]
        for (int idx = tid; idx < RCH * (KMAX / CH); idx += 256) {
            const int r = idx / (KMAX / CH), c = idx % (KMAX / CH);
            const int k0 = c * CH;
            const long row = r0 + r;
            alignas(16) T tmp[CH];
            for (int i = 0; i < CH; ++i) tmp[i] = (T)0.f;
            if (row < p.R) {
                // split-source rows: [x2 row (x_k0) | X row (K - x_k0)];
                // x_k0 is CH-aligned (binding contract), so a chunk never
                // straddles the boundary
                const T* xs = X;
                long xb = row * (long)p.K + k0;
                if (p.x2) {
                    if (k0 < p.x_k0) { xs = (const T*)p.x2; xb = row * (long)p.x_k0 + k0; }
                    else xb = row * (long)(p.K - p.x_k0) + (k0 - p.x_k0);
                }
                if (p.x_vec && k0 + CH <= p.K)
                    *(Chunk16*)tmp = *(const Chunk16*)&xs[xb];
                else
                    for (int i = 0; i < CH; ++i)
                        if (k0 + i < p.K) tmp[i] = xs[xb + i];
            }
            // lane-rotated write order: consecutive lanes write different
            // LDS rows per instruction (banks 4*{0..7}, no 16-way conflict)
#pragma unroll
            for (int i = 0; i < CH; ++i) {
                const int j = (i + tid) % CH;
                ldsXT[k0 + j][r] = tmp[j];
            }
        }
        // stage Y chunk transposed: ldsYT[n][r]
        for (int idx = tid; idx < RCH * (NMAX / CH); idx += 256) {
            const int r = idx / (NMAX / CH), c = idx % (NMAX / CH);
            const int n0 = c * CH;
            const long row = r0 + r;
            alignas(16) T tmp[CH];
            for (int i = 0; i < CH; ++i) tmp[i] = (T)0.f;
            if (row < p.R) {
                if constexpr (YF8) {
                    const unsigned char* Y8p = (const unsigned char*)p.Y;
                    long y8b = row * (long)p.N + n0;
                    if (p.y2) {  // split fp8 rows [y2 (y_k0) | Y (N - y_k0)]
                        if (n0 < p.y_k0) {
                            Y8p = (const unsigned char*)p.y2;
                            y8b = row * (long)p.y_k0 + n0;
                        } else {
                            y8b = row * (long)(p.N - p.y_k0) + (n0 - p.y_k0);
                        }
                    }
                    if (p.y_vec && n0 + CH <= p.N) {
                        // CH(=8) fp8 bytes -> 8 bf16 via packed converts
                        unsigned long long raw;
                        __builtin_memcpy(&raw, &Y8p[y8b], 8);
                        const int lo = (int)raw, hi = (int)(raw >> 32);
                        const f32x2 f0 = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
                        const f32x2 f1 = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
                        const f32x2 f2 = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
                        const f32x2 f3 = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
                        tmp[0] = (T)f0[0]; tmp[1] = (T)f0[1];
                        tmp[2] = (T)f1[0]; tmp[3] = (T)f1[1];
                        tmp[4] = (T)f2[0]; tmp[5] = (T)f2[1];
                        tmp[6] = (T)f3[0]; tmp[7] = (T)f3[1];
                    } else {
                        for (int i = 0; i < CH; ++i)
                            if (n0 + i < p.N) {
                                const int b = Y8p[y8b + i];
                                tmp[i] = (T)__builtin_amdgcn_cvt_pk_f32_fp8(b, false)[0];
                            }
                    }
                } else {
                    const T* ys = Y;
                    long yb = row * (long)p.N + n0;
                    if (p.y2) {  // [y2 row (y_k0) | Y row (N - y_k0)]
                        if (n0 < p.y_k0) { ys = (const T*)p.y2; yb = row * (long)p.y_k0 + n0; }
                        else yb = row * (long)(p.N - p.y_k0) + (n0 - p.y_k0);
                    }
                    if (p.y_vec && n0 + CH <= p.N)
                        *(Chunk16*)tmp = *(const Chunk16*)&ys[yb];
                    else
                        for (int i = 0; i < CH; ++i)
                            if (n0 + i < p.N) tmp[i] = ys[yb + i];
                }
            }
#pragma unroll
            for (int i = 0; i < CH; ++i) {
                const int j = (i + tid) % CH;
                ldsYT[n0 + j][r] = tmp[j];
            }
        }
        if (XV) {
            for (int r = tid; r < RCH; r += 256) {
                const long row = r0 + r;
                ldsXV[r] = (row < p.R) ? XV[row * p.xv_stride + p.xv_off] : (T)0.f;
            }
        }
        __syncthreads();

        for (int kk = 0; kk < RCH; kk += MT::MFMA_K) {
            typename MT::frag_t af[AKF], bf[ANF];
#pragma unroll
            for (int kf = 0; kf < AKF; ++kf)
                af[kf] = *(const typename MT::frag_t*)
                    &ldsXT[wk * AKF * 16 + kf * 16 + lrow][kk + kgrp * MT::FRAG_ELEMS];
#pragma unroll
            for (int nf = 0; nf < ANF; ++nf)
                bf[nf] = *(const typename MT::frag_t*)
                    &ldsYT[wn * ANF * 16 + nf * 16 + lrow][kk + kgrp * MT::FRAG_ELEMS];
#pragma unroll
            for (int kf = 0; kf < AKF; ++kf)
#pragma unroll
                for (int nf = 0; nf < ANF; ++nf)
                    acc[kf][nf] = MT::mfma(af[kf], bf[nf], acc[kf][nf]);
        }

        if (p.colsum || XV) {
            // split the row range across thread groups of KMAX: thread t
            // handles k = t % KMAX over its group's row slice — keeps (most of)
            // the 256 threads busy; partials combine in the final atomics
            constexpr int NSPLIT = (256 / KMAX >= 2) ? 2 : 1;
            const int k = tid % KMAX;
            const int part = tid / KMAX;
            if (k < p.K && part < NSPLIT) {
                float c1 = 0.f, x1 = 0.f;
                for (int r = part * (RCH / NSPLIT); r < (part + 1) * (RCH / NSPLIT); ++r) {
                    const float v = to_f32(ldsXT[k][r]);
                    c1 += v;
                    if (XV) x1 += v * to_f32(ldsXV[r]);
                }
                cs += c1;
                xd += x1;
            }
        }
        __syncthreads();
    }

    // epilogue: atomic accumulation (fast path) or per-block workspace rows
    // (deterministic path — summed order-independently by the caller)
    float* out_row = p.det ? p.out + (long)blockIdx.x * p.K * p.N : p.out;
#pragma unroll
    for (int kf = 0; kf < AKF; ++kf) {
#pragma unroll
        for (int nf = 0; nf < ANF; ++nf) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int k = wk * AKF * 16 + kf * 16 + kgrp * 4 + r;
                const int n = wn * ANF * 16 + nf * 16 + lrow;
                if (k < p.K && n < p.N) {
                    if (p.det) out_row[(long)k * p.N + n] = acc[kf][nf][r];
                    else atomicAdd(&out_row[(long)k * p.N + n], acc[kf][nf][r]);
                }
            }
        }
    }
    {
        constexpr int NSPLIT = (256 / KMAX >= 2) ? 2 : 1;
        const int k = tid % KMAX;
        const int part = tid / KMAX;
        if (k < p.K && part < NSPLIT) {
            if (p.det) {
                // two row-range partials per k land in separate workspace halves
                if (p.colsum)
                    p.colsum[((long)blockIdx.x * NSPLIT + part) * p.K + k] = cs;
                if (XV)
                    p.xdot[((long)blockIdx.x * NSPLIT + part) * p.K + k] = xd;
            } else {
                if (p.colsum) atomicAdd(&p.colsum[k], cs);
                if (XV) atomicAdd(&p.xdot[k], xd);
            }
        }
    }
}

extern "C" long red_gemm_nblocks(long R) {
    long chunks = (R + 63) / 64;
    return chunks < 2432 ? chunks : 2432;
}

extern "C" void red_gemm_launch(RedGemmParams p, int is_f32, hipStream_t s) {
    long blocks = red_gemm_nblocks(p.R);
    dim3 grid((unsigned)blocks), block(256);
#define DISPATCH(TT, YF)                                                    \
    do {                                                                    \
        if (p.K <= 96 && p.N <= 96)                                         \
            red_gemm_kernel<TT, 3, 3, YF><<<grid, block, 0, s>>>(p);        \
        else if (p.K <= 128 && p.N <= 32)                                   \
            red_gemm_kernel<TT, 4, 1, YF><<<grid, block, 0, s>>>(p);        \
        else if (p.K <= 160 && p.N <= 160)                                  \
            red_gemm_kernel<TT, 5, 5, YF><<<grid, block, 0, s>>>(p);        \
        else {                                                              \
            fprintf(stderr, "red_gemm: K=%d N=%d unsupported\n", p.K, p.N); \
            abort();                                                        \
        }                                                                   \
    } while (0)
    if (p.y_fp8) DISPATCH(__bf16, true);
    else if (!is_f32) DISPATCH(__bf16, false);
    else DISPATCH(float, false);
#undef DISPATCH
}
