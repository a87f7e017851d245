// Fused LSTM cell (gfx950) — one kernel per timestep over R = batch*N^2
// independent sequences with input_dim = 1 (the MPGCN temporal encoder,
// reference MPGCN.py:69,103: nn.LSTM(input=1, hidden=32, layers=1)).
//
// Forward step fuses: 4-gate GEMM (h_prev @ W_hh^T via MFMA, W_hh staged in
// LDS), the scalar-input term x_t * w_ih, bias, sigmoid/tanh nonlinearities,
// the c/h state update, and the save of post-activation gates for backward —
// one global read of h_prev/c_prev and one write of h/c/gates per step, no
// intermediate gate tensor round-trip through HBM.
//
// Gate order matches torch (i, f, g, o); W_hh is passed in its native torch
// layout (4H, H), which is exactly the transposed ([n][k]) image the MFMA
// B-fragment wants. Requires H % 16 == 0 and 4H <= 128 (H <= 32); other
// configs take the eager fallback path in Python.
#include "common.hpp"
#include "params.hpp"

template <typename T, int H>
__launch_bounds__(256) __global__ void lstm_step_fwd_kernel(LstmStepParams p) {
    using MT = MfmaTraits<T>;
    constexpr int G4 = 4 * H;
    constexpr int KP = H + MfmaTraits<T>::LDS_PAD;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    T* ldsW = (T*)smem;  // [4H][H + PAD]

    const T* __restrict__ Whh = (const T*)p.whh;
    const T* __restrict__ Hp = (const T*)p.h_prev;
    const T* __restrict__ Xp = (const T*)p.x;
    T* __restrict__ Ho = (T*)p.h_out;
    T* __restrict__ Go = (T*)p.gates_out;

    const int tid = threadIdx.x;
    for (int idx = tid; idx < G4 * KP; idx += 256) {
        const int n = idx / KP, k = idx % KP;
        ldsW[idx] = (k < H) ? Whh[(long)n * H + k] : (T)0.f;
    }
    __syncthreads();

    const int w = tid / WAVE, lane = tid % WAVE;
    const int lrow = lane & 15, kgrp = lane >> 4;
    constexpr int nf16 = G4 / 16;   // column fragments (<= 8)
    constexpr int h16 = H / 16;     // fragments per gate
    constexpr int kfrags = (H + MT::MFMA_K - 1) / MT::MFMA_K;

    for (long r0 = (long)blockIdx.x * 64 + w * 16; r0 < p.R;
         r0 += (long)gridDim.x * 64) {
        f32x4 acc[nf16] = {};
#pragma unroll
        for (int kf = 0; kf < kfrags; ++kf) {
            const int k = kf * MT::MFMA_K + kgrp * MT::FRAG_ELEMS;
            const long row = r0 + lrow;
            typename MT::frag_t af;
            if (row < p.R && k + MT::FRAG_ELEMS <= H) {
                af = *(const typename MT::frag_t*)&Hp[row * H + k];
            } else {
                alignas(16) T tmp[MT::FRAG_ELEMS];
#pragma unroll
                for (int i = 0; i < MT::FRAG_ELEMS; ++i)
                    tmp[i] = (row < p.R && k + i < H) ? Hp[row * H + k + i] : (T)0.f;
                af = *(const typename MT::frag_t*)tmp;
            }
#pragma unroll
            for (int nf = 0; nf < nf16; ++nf) {
                const typename MT::frag_t bf = *(const typename MT::frag_t*)
                    &ldsW[(nf * 16 + lrow) * KP + kf * MT::MFMA_K +
                          kgrp * MT::FRAG_ELEMS];
                acc[nf] = MT::mfma(af, bf, acc[nf]);
            }
        }

        // Per lane: rows m = r0 + kgrp*4 + r (r = 0..3), columns n = nf*16 + lrow.
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const long m = r0 + kgrp * 4 + r;
            if (m >= p.R) continue;
            const float xv = to_f32(Xp[m * p.x_stride + p.x_off]);
            // activate all gate fragments for this row, keep in registers
            float gv[nf16];
#pragma unroll
            for (int nf = 0; nf < nf16; ++nf) {
                const int n = nf * 16 + lrow;
                float v = acc[nf][r] + xv * p.wih[n] + p.bias[n];
                const int gate = n / H;
                v = (gate == 2) ? fast_tanh(v) : fast_sigmoid(v);
                gv[nf] = v;
                Go[m * G4 + n] = from_f32<T>(v);
            }
            // state update per hidden index j = jf*16 + lrow
#pragma unroll
            for (int jf = 0; jf < h16; ++jf) {
                const int j = jf * 16 + lrow;
                const float i_g = gv[0 * h16 + jf], f_g = gv[1 * h16 + jf];
                const float g_g = gv[2 * h16 + jf], o_g = gv[3 * h16 + jf];
                const float c_new = f_g * p.c_prev[m * H + j] + i_g * g_g;
                p.c_out[m * H + j] = c_new;
                Ho[m * H + j] = from_f32<T>(o_g * fast_tanh(c_new));
            }
        }
    }
}

template <typename T>
__launch_bounds__(256) __global__ void lstm_step_bwd_kernel(LstmBwdParams p) {
    const int H = p.H, G4 = 4 * H;
    const T* __restrict__ dh = (const T*)p.dh;
    const T* __restrict__ gates = (const T*)p.gates;
    T* __restrict__ dgates = (T*)p.dgates;
    const long total = p.R * H;
    for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
         idx += (long)gridDim.x * blockDim.x) {
        const long r = idx / H;
        const int j = (int)(idx % H);
        const float i_g = to_f32(gates[r * G4 + 0 * H + j]);
        const float f_g = to_f32(gates[r * G4 + 1 * H + j]);
        const float g_g = to_f32(gates[r * G4 + 2 * H + j]);
        const float o_g = to_f32(gates[r * G4 + 3 * H + j]);
        const float cv = p.c[idx];
        const float tc = fast_tanh(cv);
        const float dhv = to_f32(dh[idx]);
        float dc = dhv * o_g * (1.f - tc * tc);
        if (p.dc_in) dc += p.dc_in[idx];
        const float di = dc * g_g, dg = dc * i_g, df = dc * p.c_prev[idx];
        const float do_ = dhv * tc;
        dgates[r * G4 + 0 * H + j] = from_f32<T>(di * i_g * (1.f - i_g));
        dgates[r * G4 + 1 * H + j] = from_f32<T>(df * f_g * (1.f - f_g));
        dgates[r * G4 + 2 * H + j] = from_f32<T>(dg * (1.f - g_g * g_g));
        dgates[r * G4 + 3 * H + j] = from_f32<T>(do_ * o_g * (1.f - o_g));
        p.dc_prev[idx] = dc * f_g;
    }
}

extern "C" void lstm_step_fwd_launch(LstmStepParams p, int is_f32, hipStream_t s) {
    const int elem = is_f32 ? 4 : 2;
    const int pad = is_f32 ? 4 : 8;
    const size_t smem = (size_t)4 * p.H * (p.H + pad) * elem + 64;  // +64 B slack for k-tail fragment reads
    long tiles = (p.R + 63) / 64;
    if (tiles > 16384) tiles = 16384;
    dim3 grid((unsigned)tiles), block(256);
    if (p.H == 32) {
        if (!is_f32)
            lstm_step_fwd_kernel<__bf16, 32><<<grid, block, smem, s>>>(p);
        else
            lstm_step_fwd_kernel<float, 32><<<grid, block, smem, s>>>(p);
    } else if (p.H == 16 && is_f32) {
        // bf16 H=16 would read past the padded LDS row (MFMA_K=32 > H+pad);
        // only the f32 (MFMA_K=4) variant is safe at H=16.
        lstm_step_fwd_kernel<float, 16><<<grid, block, smem, s>>>(p);
    } else {
        fprintf(stderr, "lstm_step_fwd: unsupported H=%d (use eager path)\n", p.H);
        abort();
    }
}

extern "C" void lstm_step_bwd_launch(LstmBwdParams p, int is_f32, hipStream_t s) {
    long blocks = (p.R * p.H + 255) / 256;
    if (blocks > 16384) blocks = 16384;
    dim3 grid((unsigned)blocks), block(256);
    if (!is_f32)
        lstm_step_bwd_kernel<__bf16><<<grid, block, 0, s>>>(p);
    else
        lstm_step_bwd_kernel<float><<<grid, block, 0, s>>>(p);
}
