// Launch-parameter structs + C launcher declarations shared between the HIP
// kernel translation units and the torch binding (ext.hip).
#pragma once

#include <hip/hip_runtime.h>

struct AxisGemmParams {
    const void* AT;
    const void* X;
    void* OUT;
    void* OUT2;  // optional second epilogue copy (fp8/bf16 twin); nullable
    const float* bias;
    const float* scale;  // optional epilogue multiplier (DEVICE pointer —
                         // fp8 gradient descale, no host sync); nullable
    int M, K, L;
    int a_div; long a_bs1, a_bs2;
    int x_div; long x_bs1, x_bs2;
    int o_div; long o_bs1, o_bs2;
    int kdiv; long k_hi, k_lo;
    long k_base;         // constant X-row offset (identity-support k skip)
    int qdiv; long q_hi;
    long o_row;
    int o_mdiv; long o_m_hi, o_m_lo, o_m_base;  // OUT row mapping (0: m*o_row)
    int ogdiv; long og_hi;
    // epilogue: v = (alpha*acc + cs_beta*csub[..]) * (*scale) + bias
    const void* CSUB;    // same element type as X; nullable
    int cs_div; long cs_bs1, cs_bs2;  // CSUB instance striding
    long cs_row;         // CSUB row stride (index = cs_base + xcol_off(q) + m*cs_row)
    float alpha;         // 0 is treated as 1 by the launchers
    float cs_beta;
    int relu;
    int bias_mod;  // bias index = bias_mod ? q % bias_mod : q
    int a_vec, x_vec;
    int tiles_l;
};

struct RowGemmParams {
    const void* X;
    const void* X2;     // optional identity-block source: row k < k0 elements
                        // come from X2 (row stride k0), k >= k0 from X (row
                        // stride K - k0). Avoids materializing the identity
                        // support slot (slot_copy) just to re-read it here.
    int k0;
    const void* W;
    void* OUT;          // bf16/f32/fp8 primary output; nullable when OUT8 set
    void* OUT8;         // optional fp8 e4m3 output = fp8(v * *q_scale); nullable
    const float* q_scale;  // device ptr, quantize multiplier for OUT8
    float* amax_out;    // device ptr, atomicMax'd with block amax(|v|); nullable
    const float* bias;
    long R;
    int K, N;
    long o_row, o_off;  // OUT element index = m * o_row + o_off + n
    int relu;
    int x_vec;
};

struct LstmStepParams {
    const void* x;
    long x_stride, x_off;
    const void* h_prev;
    const float* c_prev;
    const void* whh;
    const float* wih;
    const float* bias;
    void* h_out;
    float* c_out;
    void* gates_out;
    long R;
    int H;
};

struct LstmBwdParams {
    const void* dh;
    const float* dc_in;
    const void* gates;
    const float* c_prev;
    const float* c;
    void* dgates;
    float* dc_prev;
    long R;
    int H;
};

struct RedGemmParams {
    const void* X;      // (R, K) row-major T
    const void* x2;     // optional identity-block source for X rows: k < x_k0
                        // from x2 (row stride x_k0), else X (stride K - x_k0)
    int x_k0;
    const void* y2;     // same for Y rows at y_k0
    int y_k0;
    const void* Y;      // (R, N) row-major T
    const void* xvec;   // per-row scalar T at xvec[r*xv_stride + xv_off]; nullable
    long xv_stride, xv_off;
    float* out;         // (K, N) f32, accumulated (caller zeroes)
    float* colsum;      // (K,) f32 or nullptr
    float* xdot;        // (K,) f32 or nullptr (requires xvec)
    long R;
    int K, N;
    int x_vec, y_vec;
    int y_fp8;          // Y operand is e4m3 (fp8-forward mode's saved U8)
    int det;            // deterministic: out/colsum/xdot are per-block
                        // workspaces (nblocks, ...) written with plain stores
};

struct LstmFusedParams {
    const void* x;  // (R, x_cols) T; chunk reads columns [x_off, x_off+T)
    const void* whh;   // (4H, H)
    const void* whh2;  // (H, 4H) = whh^T (backward only)
    const float* wih;  // (4H)
    const float* bias; // (4H)
    void* h_out;       // (R, H) — forward output h at chunk end
    const void* dh;    // (R, H) — backward input dL/dh(chunk end)
    float* ws_dw;      // (nblocks, 4H, H) f32 workspace
    float* ws_db;      // (nblocks, 4H)
    float* ws_dwih;    // (nblocks, 4H)
    void* dx;          // (R, x_cols) or nullptr (chunk writes its columns)
    long R;
    int T;             // chunk length (<= 8)
    // ---- chunked T > 8 support (boundary states; all nullable) ----
    long x_cols, x_off;     // x row stride / this chunk's first column
    const void* h_in;       // (R, H) bf16 state entering the chunk (null = 0)
    const float* c_in;      // (R, H) f32
    float* c_out;           // (R, H) f32 chunk-end cell state (fwd checkpoint)
    const float* dc_in;     // (R, H) f32 dL/dc(chunk end) (bwd chaining)
    void* dh_out;           // (R, H) bf16 dL/dh(chunk start)
    float* dc_out;          // (R, H) f32 dL/dc(chunk start)
};

struct ReluBwdParams {
    const void* dH;  // (R, H) bf16
    const void* Y;   // (R, H) bf16 (forward output; ignored when !mask)
    void* dY;        // (R, H) bf16 out
    void* dY8;       // optional fp8 out = fp8(dY * *q_scale); nullable
    const float* q_scale;  // device ptr (fp8 gradient quantize scale)
    float* amax_out; // device ptr, atomicMax'd with amax(|dY|); nullable
    float* colsum;   // (H,) f32 zeroed, or (nblocks, H) workspace when det
    long total;      // R * H (multiple of 8)
    int H;           // power of two dividing 2048
    int mask;
    int det;
};

extern "C" {
void slab_colsum_launch(const float* ws, float* out, long nb, long E, hipStream_t s);
void axis_gemm_launch(AxisGemmParams p, int instances, int is_f32, hipStream_t s);
void axis_gemm_fp8_launch(AxisGemmParams p, int instances, int out_kind, hipStream_t s);
void red_gemm_launch(RedGemmParams p, int is_f32, hipStream_t s);
void row_gemm_launch(RowGemmParams p, int is_f32, hipStream_t s);
void row_gemm_fp8_launch(RowGemmParams p, hipStream_t s);
void lstm_step_fwd_launch(LstmStepParams p, int is_f32, hipStream_t s);
void lstm_step_bwd_launch(LstmBwdParams p, int is_f32, hipStream_t s);
void lstm_fused_fwd_launch(LstmFusedParams p, hipStream_t s);
void lstm_fused_bwd_launch(LstmFusedParams p, hipStream_t s);
int lstm_fused_bwd_blocks(long R);
void relu_bwd_colsum_launch(ReluBwdParams p, hipStream_t s);
long relu_bwd_nblocks(long total);
long red_gemm_nblocks(long R);
void fp8_scale_update_launch(float* amax, float* scale, float* inv,
                             float margin, hipStream_t s);
void adam_flat_launch(float* p, const float* g, float* m, float* v, float* t,
                      long E, float lr, float b1, float b2, float eps, float wd,
                      hipStream_t s);
void slot_copy_launch(const void* src, void* dst, long rows, int row_b,
                      long dst_stride_b, hipStream_t s);
void slab_colsum3_launch(const float* w1, float* o1, long E1,
                         const float* w2, float* o2, long E2,
                         const float* w3, float* o3, long E3,
                         long nb, hipStream_t s);
void rwd_rowsum_launch(const float* A, float* d, long rows, int N, hipStream_t s);
void rwd_colsum_launch(const float* A, float* dc, long B, int N, hipStream_t s);
void dual_bwd_norm_launch(const float* A, const float* dc, float* OUT, long B,
                          int N, long sOUT, hipStream_t s);
void localpool_seed_launch(const float* A, const float* d, float* OUT,
                           long B, int N, long sOUT, hipStream_t s);
void cheby_seed_launch(const float* A, const float* d, float* OUT, long B,
                       int N, long sOUT, float lam, hipStream_t s);
void rwd_norm_t_launch(const float* A, const float* d, float* OUT, long B,
                       int N, long sOUT, hipStream_t s);
}
