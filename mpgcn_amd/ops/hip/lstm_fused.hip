// Fully-fused register-resident LSTM (gfx950, bf16, H = 32, T <= 8).
//
// The MPGCN temporal encoder runs R = batch*N^2 independent scalar-input
// sequences of length T <= 8 and only the LAST hidden state is consumed
// (MPGCN.py:103-104). That makes the whole sequence small enough to live in
// registers per 16-row wave tile:
//
//   forward:  h/c stay in registers across all T steps; the ONLY global
//             traffic is x (R*T scalars) in and h_T out — no per-step state
//             round-trips through HBM (the slab variant in lstm.hip moves
//             ~9 GB/step at the flagship config; this moves ~0.2 GB).
//   backward: recomputes the forward states in registers (bit-identical MFMA
//             sequence), then walks t = T-1..0 computing the gate gradients,
//             the dh chain (MFMA vs W_hh), and the dW_hh / dbias / dw_ih
//             partials in-register; partials land in a per-block f32
//             workspace reduced by the fixed-order slab_colsum kernel
//             (elemwise.hip) — no atomics anywhere, so gradients are
//             bitwise-reproducible; no dgates materialization.
//
// Weight-grad MFMAs pair two timesteps per K=32 contraction (16 rows each).
// Per-wave LDS images: dg_img [2*16][4H] (dh A-operand, vector reads; dW
// A-operand via strided scalar reads), hT_img [H][2*16] (dW B-operand,
// vector reads). W_hh staged twice: [4H][H] (gate GEMM B) and [H][4H]
// (dh-chain B).
#include "common.hpp"
#include "params.hpp"

#define LF_H 32
#define LF_G4 128
// per-wave image paddings
#define DG_LD (LF_G4 + 8)  // dg_img row length (rows = 32 paired rows)
#define HT_LD (32 + 8)     // hT_img row length (rows = 32 k values)


typedef __attribute__((ext_vector_type(8))) __bf16 lf_frag;

// Wave-local LDS fence: the transpose tiles are PER-WAVE, so cross-lane
// visibility needs only this wave's ds ops committed (lgkmcnt(0)), not a
// block barrier — at 1 block/CU a __syncthreads here would serialize all
// four independent waves on every step.
__device__ __forceinline__ void lds_wave_fence() {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

#define LF_XCOLS 8  // x is padded to 8 columns so each row is one 16-B load

// gate activation order: i, f, g, o (torch chunk order)
__device__ __forceinline__ float lf_act(float v, int gate) {
    return (gate == 2) ? fast_tanh(v) : fast_sigmoid(v);
}

// one gate GEMM: acc[nf] = h_frag @ whhT fragments (8 MFMAs, K = 32)
__device__ __forceinline__ void lf_gate_mfma(const __bf16* ldsW, lf_frag h_frag,
                                             int lrow, int kgrp, f32x4 acc[8]) {
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
        const lf_frag bf = *(const lf_frag*)
            &ldsW[(nf * 16 + lrow) * (LF_H + 8) + kgrp * 8];
        acc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(h_frag, bf, acc[nf], 0, 0, 0);
    }
}

template <int T>
__launch_bounds__(256) __global__ void lstm_fused_fwd_kernel(LstmFusedParams p) {
    __shared__ __bf16 ldsW[LF_G4 * (LF_H + 8)];
    __shared__ __bf16 ldsH[4][16 * (LF_H + 8)];  // per-wave h transpose tile

    const __bf16* __restrict__ X = (const __bf16*)p.x;
    const __bf16* __restrict__ Whh = (const __bf16*)p.whh;
    __bf16* __restrict__ Ho = (__bf16*)p.h_out;

    const int tid = threadIdx.x;
    for (int i = tid; i < LF_G4 * (LF_H + 8); i += 256) {
        const int n = i / (LF_H + 8), k = i % (LF_H + 8);
        ldsW[i] = (k < LF_H) ? Whh[n * LF_H + k] : (__bf16)0.f;
    }
    __syncthreads();

    const int w = tid / WAVE, lane = tid % WAVE;
    const int lrow = lane & 15, kgrp = lane >> 4;
    __bf16* myH = &ldsH[w][0];
    // hoist the per-lane w_ih/bias values out of the step loops (they are
    // otherwise re-read from global memory every step)
    float wih_r[8], bias_r[8];
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
        wih_r[nf] = p.wih[nf * 16 + lrow];
        bias_r[nf] = p.bias[nf * 16 + lrow];
    }

    const __bf16* __restrict__ Hin = (const __bf16*)p.h_in;
    const float* __restrict__ Cin = p.c_in;
    float* __restrict__ Cout = p.c_out;

    const long ntiles = (p.R + 63) / 64;
    for (long tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const long r0 = tile * 64 + w * 16;
        alignas(16) __bf16 xrow[4][LF_XCOLS];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const long m = r0 + kgrp * 4 + r;
            if (m < p.R)
                *(Chunk16*)&xrow[r][0] = *(const Chunk16*)&X[m * p.x_cols + p.x_off];
            else for (int t2 = 0; t2 < LF_XCOLS; ++t2) xrow[r][t2] = (__bf16)0.f;
        }
        // chunk-entry state: zeros for the first chunk, the previous chunk's
        // boundary checkpoint otherwise (chunked T > 8 schedule)
        lf_frag h_frag = {};
        if (Hin) {
            const long row = r0 + lrow;
            if (row < p.R) h_frag = *(const lf_frag*)&Hin[row * LF_H + kgrp * 8];
        }
        float c[2][4] = {};  // c[jf][r] for j = jf*16 + lrow, row = r0 + kgrp*4 + r
        if (Cin) {
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long m = r0 + kgrp * 4 + r;
                    if (m < p.R) c[jf][r] = Cin[m * LF_H + jf * 16 + lrow];
                }
        }
#pragma unroll
        for (int t = 0; t < T; ++t) {
            f32x4 acc[8] = {};
            lf_gate_mfma(ldsW, h_frag, lrow, kgrp, acc);
            // activations + state update (C-layout)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const float xv = to_f32(xrow[r][t]);
                float gv[8];
#pragma unroll
                for (int nf = 0; nf < 8; ++nf)
                    gv[nf] = lf_act(acc[nf][r] + xv * wih_r[nf] + bias_r[nf],
                                    (nf * 16 + lrow) / LF_H);
#pragma unroll
                for (int jf = 0; jf < 2; ++jf) {
                    const float c_new = gv[2 + jf] * c[jf][r] + gv[0 + jf] * gv[4 + jf];
                    c[jf][r] = c_new;
                    // h in C-layout -> per-wave LDS tile for transposition
                    myH[(kgrp * 4 + r) * (LF_H + 8) + jf * 16 + lrow] =
                        (__bf16)(gv[6 + jf] * fast_tanh(c_new));
                }
            }
            lds_wave_fence();
            h_frag = *(const lf_frag*)&myH[lrow * (LF_H + 8) + kgrp * 8];
            lds_wave_fence();
        }
        // store chunk-end h, vectorized: lane holds row lrow's k-run
        const long row = r0 + lrow;
        if (row < p.R) *(lf_frag*)&Ho[row * LF_H + kgrp * 8] = h_frag;
        if (Cout) {
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long m = r0 + kgrp * 4 + r;
                    if (m < p.R) Cout[m * LF_H + jf * 16 + lrow] = c[jf][r];
                }
        }
    }
}

// NOTE on gv indexing above: gate order i,f,g,o over n = 0..127 means
// nf 0..1 = i, 2..3 = f, 4..5 = g, 6..7 = o; jf = n/16 % 2 selects the half.

template <int T, bool CH = false>
__launch_bounds__(256) __global__ void lstm_fused_bwd_kernel(LstmFusedParams p) {
    __shared__ __bf16 ldsW[LF_G4 * (LF_H + 8)];   // [n][k] gate-GEMM B image
    __shared__ __bf16 ldsW2[LF_H * (LF_G4 + 8)];  // [k][n] dh-chain B image
    __shared__ __bf16 ldsDG[4][32 * DG_LD];       // per-wave paired dgates [row'][n]
    __shared__ __bf16 ldsHT[4][LF_H * HT_LD];     // per-wave paired h_prev^T [k][row']
    // c states live in LDS (they would otherwise cost 8*T VGPRs per lane):
    // CST[w][t][row][j], row stride 33 breaks write conflicts
    __shared__ float ldsC[4][T * 16 * 33];

    const __bf16* __restrict__ X = (const __bf16*)p.x;
    const __bf16* __restrict__ Whh = (const __bf16*)p.whh;
    const __bf16* __restrict__ Whh2 = (const __bf16*)p.whh2;
    const __bf16* __restrict__ DH = (const __bf16*)p.dh;
    __bf16* __restrict__ DX = (__bf16*)p.dx;

    const int tid = threadIdx.x;
    for (int i = tid; i < LF_G4 * (LF_H + 8); i += 256) {
        const int n = i / (LF_H + 8), k = i % (LF_H + 8);
        ldsW[i] = (k < LF_H) ? Whh[n * LF_H + k] : (__bf16)0.f;
    }
    for (int i = tid; i < LF_H * (LF_G4 + 8); i += 256) {
        const int k = i / (LF_G4 + 8), n = i % (LF_G4 + 8);
        ldsW2[i] = (n < LF_G4) ? Whh2[k * LF_G4 + n] : (__bf16)0.f;
    }
    __syncthreads();

    const int w = tid / WAVE, lane = tid % WAVE;
    const int lrow = lane & 15, kgrp = lane >> 4;
    __bf16* myDG = &ldsDG[w][0];
    __bf16* myHT = &ldsHT[w][0];
    __bf16* myH = &ldsHT[w][0];  // phase-A transpose scratch aliases the hT image
    float* myC = &ldsC[w][0];
    float wih_r[8], bias_r[8];
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
        wih_r[nf] = p.wih[nf * 16 + lrow];
        bias_r[nf] = p.bias[nf * 16 + lrow];
    }

    // per-wave dW accumulator: D[i=n][j=k], 8 m-frags x 2 j-frags
    f32x4 dw_acc[8][2] = {};
    float db_acc[8] = {}, dwih_acc[8] = {};

    // CH=false (the whole-sequence T <= 8 path, incl. the flagship T=7)
    // compiles the chunk-boundary logic OUT — the extra live state pushed
    // the T=7 instantiation into VGPR spills otherwise.
    const __bf16* __restrict__ Hin = CH ? (const __bf16*)p.h_in : nullptr;
    const float* __restrict__ Cin = CH ? p.c_in : nullptr;
    const float* __restrict__ DCin = CH ? p.dc_in : nullptr;
    __bf16* __restrict__ DHout = CH ? (__bf16*)p.dh_out : nullptr;
    float* __restrict__ DCout = CH ? p.dc_out : nullptr;

    const long ntiles = (p.R + 63) / 64;
    for (long tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const long r0 = tile * 64 + w * 16;
        alignas(16) __bf16 xrow[4][LF_XCOLS];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const long m = r0 + kgrp * 4 + r;
            if (m < p.R)
                *(Chunk16*)&xrow[r][0] = *(const Chunk16*)&X[m * p.x_cols + p.x_off];
            else for (int t2 = 0; t2 < LF_XCOLS; ++t2) xrow[r][t2] = (__bf16)0.f;
        }
        // chunk-entry state (chunked T > 8: previous chunk's checkpoint)
        lf_frag h_in_frag = {};
        float c0r[2][4] = {};  // initial c (phase B's c_prev at t = 0)
        if (CH && Hin) {
            const long row = r0 + lrow;
            if (row < p.R)
                h_in_frag = *(const lf_frag*)&Hin[row * LF_H + kgrp * 8];
        }
        if (CH && Cin) {
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long m = r0 + kgrp * 4 + r;
                    if (m < p.R) c0r[jf][r] = Cin[m * LF_H + jf * 16 + lrow];
                }
        }
        // ---- phase A: forward recompute; h states in registers, c in LDS ----
        lf_frag h_states[T];  // h AFTER step t (A-frag layout)
#define CST(t, r, jf) myC[(t) * 16 * 33 + (kgrp * 4 + (r)) * 33 + (jf) * 16 + lrow]
        {
            lf_frag h_frag = h_in_frag;
            float c[2][4];
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
#pragma unroll
                for (int r = 0; r < 4; ++r) c[jf][r] = c0r[jf][r];
#pragma unroll
            for (int t = 0; t < T; ++t) {
                f32x4 acc[8] = {};
                lf_gate_mfma(ldsW, h_frag, lrow, kgrp, acc);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const float xv = to_f32(xrow[r][t]);
                    float gv[8];
#pragma unroll
                    for (int nf = 0; nf < 8; ++nf)
                        gv[nf] = lf_act(acc[nf][r] + xv * wih_r[nf] + bias_r[nf],
                                        (nf * 16 + lrow) / LF_H);
#pragma unroll
                    for (int jf = 0; jf < 2; ++jf) {
                        const float c_new = gv[2 + jf] * c[jf][r] + gv[0 + jf] * gv[4 + jf];
                        c[jf][r] = c_new;
                        CST(t, r, jf) = c_new;
                        myH[(kgrp * 4 + r) * (LF_H + 8) + jf * 16 + lrow] =
                            (__bf16)(gv[6 + jf] * fast_tanh(c_new));
                    }
                }
                lds_wave_fence();
                h_frag = *(const lf_frag*)&myH[lrow * (LF_H + 8) + kgrp * 8];
                lds_wave_fence();
                h_states[t] = h_frag;
            }
        }

        // ---- phase B: backward t = T-1 .. 0 ----
        // dh, dc in C-layout registers
        float dh[2][4], dc[2][4] = {};
#pragma unroll
        for (int jf = 0; jf < 2; ++jf)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const long m = r0 + kgrp * 4 + r;
                dh[jf][r] = (m < p.R) ? to_f32(DH[m * LF_H + jf * 16 + lrow]) : 0.f;
                if (CH && DCin && m < p.R) dc[jf][r] = DCin[m * LF_H + jf * 16 + lrow];
            }

#pragma unroll
        for (int tt = 0; tt < T; ++tt) {
            const int t = T - 1 - tt;
            const int slot = tt % 2;
            // recompute this step's gates from h_prev
            lf_frag hp = h_in_frag;
            if (t > 0) hp = h_states[t - 1];
            f32x4 acc[8] = {};
            lf_gate_mfma(ldsW, hp, lrow, kgrp, acc);

            float dgp[8][4];  // pre-activation gate grads, C-layout
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const long m = r0 + kgrp * 4 + r;
                const bool ok = m < p.R;
                const float xv = to_f32(xrow[r][t]);
                float gv[8];
#pragma unroll
                for (int nf = 0; nf < 8; ++nf)
                    gv[nf] = lf_act(acc[nf][r] + xv * wih_r[nf] + bias_r[nf],
                                    (nf * 16 + lrow) / LF_H);
#pragma unroll
                for (int jf = 0; jf < 2; ++jf) {
                    const float i_g = gv[0 + jf], f_g = gv[2 + jf];
                    const float g_g = gv[4 + jf], o_g = gv[6 + jf];
                    const float c_t = CST(t, r, jf);
                    const float c_prev = (t > 0) ? CST(t - 1, r, jf) : c0r[jf][r];
                    const float tc = fast_tanh(c_t);
                    float d_c = dc[jf][r] + dh[jf][r] * o_g * (1.f - tc * tc);
                    const float d_i = d_c * g_g, d_g = d_c * i_g, d_f = d_c * c_prev;
                    const float d_o = dh[jf][r] * tc;
                    const float da_i = ok ? d_i * i_g * (1.f - i_g) : 0.f;
                    const float da_f = ok ? d_f * f_g * (1.f - f_g) : 0.f;
                    const float da_g = ok ? d_g * (1.f - g_g * g_g) : 0.f;
                    const float da_o = ok ? d_o * o_g * (1.f - o_g) : 0.f;
                    dgp[0 + jf][r] = da_i;
                    dgp[2 + jf][r] = da_f;
                    dgp[4 + jf][r] = da_g;
                    dgp[6 + jf][r] = da_o;
                    dc[jf][r] = d_c * f_g;
                    db_acc[0 + jf] += da_i; db_acc[2 + jf] += da_f;
                    db_acc[4 + jf] += da_g; db_acc[6 + jf] += da_o;
                    dwih_acc[0 + jf] += da_i * xv; dwih_acc[2 + jf] += da_f * xv;
                    dwih_acc[4 + jf] += da_g * xv; dwih_acc[6 + jf] += da_o * xv;
                }
            }

            // optional dx[row, t] = sum_n dgp[row, n] * wih[n]
            if (DX) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    float part = 0.f;
#pragma unroll
                    for (int nf = 0; nf < 8; ++nf)
                        part += dgp[nf][r] * wih_r[nf];
                    // reduce over the 16 lrow lanes (same row across lrow)
#pragma unroll
                    for (int s = 1; s < 16; s <<= 1)
                        part += __shfl_xor(part, s);
                    const long m = r0 + kgrp * 4 + r;
                    if (lrow == 0 && m < p.R)
                        DX[m * p.x_cols + p.x_off + t] = (__bf16)part;
                }
            }

            // stage dgates [row'][n] and h_prev^T [k][row'] into this slot
#pragma unroll
            for (int r = 0; r < 4; ++r)
#pragma unroll
                for (int nf = 0; nf < 8; ++nf)
                    myDG[(slot * 16 + kgrp * 4 + r) * DG_LD + nf * 16 + lrow] =
                        (__bf16)dgp[nf][r];
            {
                // hp is A-frag layout: lane holds h_prev[row=lrow][k=kgrp*8+j]
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    myHT[(kgrp * 8 + j) * HT_LD + slot * 16 + lrow] = hp[j];
            }
            lds_wave_fence();

            // dh chain: dh_prev[row][k] = sum_n dgp[row][n] * Whh[n][k]
            // (also at t = 0 when the chunk must emit d(h_in) for the
            // previous chunk's backward)
            if (t > 0 || DHout) {
                f32x4 dh_acc[2] = {};
#pragma unroll
                for (int kf = 0; kf < 4; ++kf) {
                    const lf_frag af = *(const lf_frag*)
                        &myDG[(slot * 16 + lrow) * DG_LD + kf * 32 + kgrp * 8];
#pragma unroll
                    for (int jf = 0; jf < 2; ++jf) {
                        const lf_frag bf = *(const lf_frag*)
                            &ldsW2[(jf * 16 + lrow) * (LF_G4 + 8) + kf * 32 + kgrp * 8];
                        dh_acc[jf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            af, bf, dh_acc[jf], 0, 0, 0);
                    }
                }
#pragma unroll
                for (int jf = 0; jf < 2; ++jf)
#pragma unroll
                    for (int r = 0; r < 4; ++r) dh[jf][r] = dh_acc[jf][r];
            }

            // dW pair MFMA when both slots are filled (or at the last step)
            if (slot == 1 || t == 0) {
                if (slot == 0) {  // odd T tail: zero the unused slot 1
#pragma unroll
                    for (int r = 0; r < 4; ++r)
#pragma unroll
                        for (int nf = 0; nf < 8; ++nf)
                            myDG[(16 + kgrp * 4 + r) * DG_LD + nf * 16 + lrow] = (__bf16)0.f;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        myHT[(kgrp * 8 + j) * HT_LD + 16 + lrow] = (__bf16)0.f;
                    lds_wave_fence();
                }
                // dW[n][k] += sum_{row'} dg[row'][n] * h_prev[row'][k]
#pragma unroll
                for (int mf = 0; mf < 8; ++mf) {
                    // A-frag via strided scalar reads of dg_img (column n)
                    alignas(16) __bf16 a_sc[8];
                    const int n = mf * 16 + lrow;
#pragma unroll
                    for (int jj = 0; jj < 8; ++jj)
                        a_sc[jj] = myDG[(kgrp * 8 + jj) * DG_LD + n];
                    const lf_frag af = *(const lf_frag*)a_sc;
#pragma unroll
                    for (int jf = 0; jf < 2; ++jf) {
                        const lf_frag bf = *(const lf_frag*)
                            &myHT[(jf * 16 + lrow) * HT_LD + kgrp * 8];
                        dw_acc[mf][jf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            af, bf, dw_acc[mf][jf], 0, 0, 0);
                    }
                }
            }
            lds_wave_fence();
        }
        // chunk-start gradients out (chained into the previous chunk's bwd)
        if (CH && DHout) {
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long m = r0 + kgrp * 4 + r;
                    if (m < p.R) {
                        DHout[m * LF_H + jf * 16 + lrow] = (__bf16)dh[jf][r];
                        DCout[m * LF_H + jf * 16 + lrow] = dc[jf][r];
                    }
                }
        }
    }
    __syncthreads();  // before reusing per-wave LDS as the block-reduce scratch

    // ---- write per-block partials to the workspace ----
    // block-level reduce across the 4 waves via LDS (reuse ldsDG as f32 scratch)
    float* red = (float*)&ldsDG[0][0];  // 4H * H floats = 16 KB
    for (int i = tid; i < LF_G4 * LF_H; i += 256) red[i] = 0.f;
    __syncthreads();
    for (int ww = 0; ww < 4; ++ww) {
        if (w == ww) {
#pragma unroll
            for (int mf = 0; mf < 8; ++mf)
#pragma unroll
                for (int jf = 0; jf < 2; ++jf)
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const int n = mf * 16 + kgrp * 4 + r;
                        const int k = jf * 16 + lrow;
                        red[n * LF_H + k] += dw_acc[mf][jf][r];
                    }
        }
        __syncthreads();
    }
    float* ws = p.ws_dw + (long)blockIdx.x * LF_G4 * LF_H;
    for (int i = tid; i < LF_G4 * LF_H; i += 256) ws[i] = red[i];

    // dbias / dwih: lane covers n = nf*16 + lrow for its 8 nf; rows are summed
    // already. Reduce over the 4 kgrp lanes sharing n with a fixed-order
    // shuffle tree, stage per-wave rows in LDS, then sum waves in fixed order
    // — bitwise-reproducible (an LDS float atomicAdd here made the whole
    // backward nondeterministic across runs).
    float* redb = (float*)&ldsHT[0][0];  // 4 waves * 2 * 4H floats
    __syncthreads();
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
        float vb = db_acc[nf], vw = dwih_acc[nf];
        vb += __shfl_down(vb, 32, 64); vb += __shfl_down(vb, 16, 64);
        vw += __shfl_down(vw, 32, 64); vw += __shfl_down(vw, 16, 64);
        if (lane < 16) {
            redb[(w * 2 + 0) * LF_G4 + nf * 16 + lane] = vb;
            redb[(w * 2 + 1) * LF_G4 + nf * 16 + lane] = vw;
        }
    }
    __syncthreads();
    for (int i = tid; i < LF_G4; i += 256) {
        float sb = 0.f, sw = 0.f;
#pragma unroll
        for (int wv = 0; wv < 4; ++wv) {
            sb += redb[(wv * 2 + 0) * LF_G4 + i];
            sw += redb[(wv * 2 + 1) * LF_G4 + i];
        }
        p.ws_db[(long)blockIdx.x * LF_G4 + i] = sb;
        p.ws_dwih[(long)blockIdx.x * LF_G4 + i] = sw;
    }
}

extern "C" void lstm_fused_fwd_launch(LstmFusedParams p, hipStream_t s) {
    long tiles = (p.R + 63) / 64;
    long blocks = tiles < 8192 ? tiles : 8192;
    dim3 grid((unsigned)blocks), block(256);
    switch (p.T) {
#define CASE(TT) case TT: lstm_fused_fwd_kernel<TT><<<grid, block, 0, s>>>(p); break;
        CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
        default: fprintf(stderr, "lstm_fused_fwd: T=%d unsupported\n", p.T); abort();
    }
}

extern "C" int lstm_fused_bwd_blocks(long R) {
    long tiles = (R + 63) / 64;
    return (int)(tiles < 512 ? tiles : 512);
}

extern "C" void lstm_fused_bwd_launch(LstmFusedParams p, hipStream_t s) {
    dim3 grid(lstm_fused_bwd_blocks(p.R)), block(256);
    const bool ch = p.h_in || p.dc_in || p.dh_out;
    switch (p.T) {
#define CASE(TT) \
    case TT: { \
        if (ch) { \
            lstm_fused_bwd_kernel<TT, true><<<grid, block, 0, s>>>(p); \
        } else { \
            lstm_fused_bwd_kernel<TT, false><<<grid, block, 0, s>>>(p); \
        } \
    } break;
        CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
        default: fprintf(stderr, "lstm_fused_bwd: T=%d unsupported\n", p.T); abort();
    }
}
