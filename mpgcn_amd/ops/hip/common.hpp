// Shared helpers for the MPGCN-MI355X HIP kernels (gfx950 / CDNA4 only).
//
// Wavefront is 64-wide; MFMA tiles are 16x16 with K=32 (bf16, f32 accumulate)
// or K=4 (exact f32). No CUDA-compat paths, no other archs.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// ---------------------------------------------------------------------------
// MFMA traits: one 16x16 output fragment per wave.
//   bf16: v_mfma_f32_16x16x32_bf16 — lane l holds A[row=l&15][k=(l>>4)*8 + j],
//         B[k=(l>>4)*8 + j][col=l&15], j = 0..7 (one bf16x8 = 4 VGPRs each).
//   f32:  v_mfma_f32_16x16x4_f32  — lane l holds A[l&15][l>>4], B[l>>4][l&15].
//   C/D (both): col = l&15, row = (l>>4)*4 + reg, reg = 0..3 (f32x4).
// ---------------------------------------------------------------------------
template <typename T>
struct MfmaTraits;

template <>
struct MfmaTraits<__bf16> {
    static constexpr int MFMA_K = 32;     // k-depth of one MFMA instruction
    static constexpr int FRAG_ELEMS = 8;  // per-lane A/B elements (contiguous in k)
    static constexpr int LDS_PAD = 8;     // pad elements per LDS row (16 B)
    using frag_t = bf16x8;
    __device__ static inline f32x4 mfma(frag_t a, frag_t b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
};

template <>
struct MfmaTraits<float> {
    static constexpr int MFMA_K = 4;
    static constexpr int FRAG_ELEMS = 1;
    static constexpr int LDS_PAD = 4;  // 16 B
    using frag_t = float;
    __device__ static inline f32x4 mfma(frag_t a, frag_t b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
    }
};

// fp8 (OCP e4m3, gfx950 native): same 16x16x32 shape as bf16 — the MFMA rate
// is identical, but operands are 1 byte, so a K-stage of equal LDS footprint
// covers 2x the K depth (the probe instantiates BK=128). Raw storage type is
// unsigned char; torch::kFloat8_e4m3fn bytes are bit-compatible.
template <>
struct MfmaTraits<unsigned char> {
    static constexpr int MFMA_K = 32;
    static constexpr int FRAG_ELEMS = 8;   // 8 fp8 bytes = one i64 operand
    static constexpr int LDS_PAD = 16;     // 16 B
    using frag_t = long;
    __device__ static inline f32x4 mfma(frag_t a, frag_t b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, c, 0, 0, 0);
    }
};

template <typename T>
__device__ __forceinline__ T from_f32(float v);
template <>
__device__ __forceinline__ __bf16 from_f32<__bf16>(float v) { return (__bf16)v; }
template <>
__device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ unsigned char from_f32<unsigned char>(float v) {
    // v_cvt_pk_fp8_f32 (OCP e4m3 on gfx950) does NOT saturate: inputs past
    // +-448 convert to NaN, which poisons fp8 gradient tensors whenever a
    // delayed scale lags a >2x amax jump (and fmaxf-based amax recording
    // silently drops NaN, locking the scale — a measured training doom
    // loop). Clamp to the e4m3 range first; fminf/fmaxf also squash NaN.
    v = fminf(fmaxf(v, -448.f), 448.f);
    return (unsigned char)(__builtin_amdgcn_cvt_pk_fp8_f32(v, 0.f, 0, false) & 0xff);
}

__device__ __forceinline__ float to_f32(__bf16 v) { return (float)v; }
__device__ __forceinline__ float to_f32(float v) { return v; }
__device__ __forceinline__ float to_f32(unsigned char v) {
    // OCP e4m3 byte -> f32 (v_cvt_pk_f32_fp8, low half)
    typedef __attribute__((ext_vector_type(2))) float _f2;
    return ((_f2)__builtin_amdgcn_cvt_pk_f32_fp8((int)v, false))[0];
}

// Fast transcendentals: v_rcp_f32-based sigmoid/tanh (no IEEE division
// sequences, no libm branches — hipcc otherwise emits v_div_scale/div_fixup
// chains and branchy tanhf that dominate the LSTM kernels' issue time).
// Accuracy ~1 ulp of rcp (~1e-7 rel) — far below bf16 resolution.
__device__ __forceinline__ float fast_sigmoid(float v) {
    return __builtin_amdgcn_rcpf(1.f + __expf(-v));
}
__device__ __forceinline__ float fast_tanh(float v) {
    // tanh(x) = 1 - 2/(exp(2x)+1); saturates correctly at +-inf
    return 1.f - 2.f * __builtin_amdgcn_rcpf(__expf(2.f * v) + 1.f);
}

// 16-byte raw copy chunk (8 bf16 / 4 f32).
struct alignas(16) Chunk16 { int v[4]; };

#define HIP_CHECK(expr)                                              \
    do {                                                             \
        hipError_t _e = (expr);                                      \
        if (_e != hipSuccess) {                                      \
            fprintf(stderr, "HIP error %s at %s:%d\n",               \
                    hipGetErrorString(_e), __FILE__, __LINE__);      \
            abort();                                                 \
        }                                                            \
    } while (0)
