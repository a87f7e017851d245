// Torch bindings for the MPGCN-MI355X HIP kernels (gfx950 only).
//
// Shape/layout contracts are documented per function; the Python wrappers in
// mpgcn_amd/ops/functional.py pre-permute the (tiny) graph operands into the
// row-major (M, K) "AT" layouts the axis_gemm kernel consumes.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "params.hpp"

namespace {

void check_in(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16 || t.scalar_type() == torch::kFloat,
                name, " must be bf16 or f32");
}

int is_f32(const torch::Tensor& t) { return t.scalar_type() == torch::kFloat ? 1 : 0; }

int chunk_elems(const torch::Tensor& t) { return is_f32(t) ? 4 : 8; }

hipStream_t stream() { return c10::hip::getCurrentHIPStream().stream(); }

const float* bias_ptr(const c10::optional<torch::Tensor>& b) {
    if (!b.has_value() || !b->defined()) return nullptr;
    TORCH_CHECK(b->scalar_type() == torch::kFloat, "bias must be f32");
    TORCH_CHECK(b->is_contiguous(), "bias must be contiguous");
    return b->data_ptr<float>();
}

}  // namespace


namespace {
// identity-slot fill via the streaming slot_copy kernel (aten's strided
// copy runs at ~1 TB/s on these shapes); falls back to aten when the row
// isn't 16-byte chunkable
void fill_slot0(torch::Tensor& dst5, const torch::Tensor& src4, long S) {
    const long rows = src4.numel() / src4.size(-1);
    const long row_b = src4.size(-1) * src4.element_size();
    if (row_b % 16 == 0) {
        slot_copy_launch(src4.data_ptr(), dst5.data_ptr(), rows, (int)row_b,
                         S * row_b, stream());
    } else {
        dst5.select(3, 0).copy_(src4);
    }
}
}  // namespace

// U[b,m,d,o,l] = sum_n GT[(b,)o,m,n] X[b,n,d,l].
// X: (B, N, N, C); GT: (S, N, N) static or (B, S, N, N) dynamic, ALREADY
// transposed per support (GT[..., m, n] = G[..., n, m]). Out: (B, N, N, S, C).
// Mode-1 (origin-axis) product: the reference's einsum('bncl,nm->bmcl') /
// ('bncl,bnm->bmcl') pair (reference MPGCN.py:30,38), batched over supports.
// Rectangular: X may be destination-sharded, (B, No, Nd, C) with Nd != No
// (the region-partition path, mpgcn_amd/parallel/region.py).
// id_skip: the FIRST support is the identity matrix (T_0 = I for every
// Chebyshev-family kernel type, GCN.py:128-138), so its product is X itself
// — GT is then the REDUCED stack WITHOUT support 0 (S-1 supports), the GEMM
// fills only slots 1..S-1 of U, and the caller copies X into slot 0
// (ops/functional.py). Cuts 1/S of the contraction's FLOPs and bytes.
torch::Tensor bdgcn_mode1(torch::Tensor X, torch::Tensor GT, bool id_skip,
                          bool no_fill) {
    check_in(X, "X");
    check_in(GT, "GT");
    const bool dyn = GT.dim() == 4;
    const long B = X.size(0), No = X.size(1), Nd = X.size(2), C = X.size(3);
    const long Se = dyn ? GT.size(1) : GT.size(0);  // supports in the GEMM
    // no_fill (id_skip only): return U with ONLY the Se computed slots — no
    // materialized identity slot. Consumers read the identity block straight
    // from X via row_gemm_split / red_gemm_split, so the slot_copy read+write
    // disappears from the step.
    const bool red = id_skip && no_fill;
    const long S = (id_skip && !red) ? Se + 1 : Se;  // slots in U
    TORCH_CHECK(GT.size(-1) == No && GT.size(-2) == No, "shape mismatch");
    TORCH_CHECK(!dyn || GT.size(0) == B, "dynamic GT batch mismatch");
    TORCH_CHECK(B * Se <= 65535, "too many instances");
    auto U = torch::empty({B, No, Nd, S, C}, X.options());

    AxisGemmParams p{};
    p.AT = GT.data_ptr();
    p.X = X.data_ptr();
    p.OUT = (id_skip && !red)
                ? (void*)((char*)U.data_ptr() + C * U.element_size())
                : U.data_ptr();
    p.M = (int)No; p.K = (int)No; p.L = (int)(Nd * C);
    p.a_div = (int)Se; p.a_bs1 = dyn ? Se * No * No : 0; p.a_bs2 = No * No;
    p.x_div = (int)Se; p.x_bs1 = No * Nd * C; p.x_bs2 = 0;
    p.o_div = (int)Se; p.o_bs1 = No * Nd * S * C; p.o_bs2 = C;
    p.kdiv = 1; p.k_lo = Nd * C;
    p.qdiv = 0;
    p.o_row = Nd * S * C;
    p.ogdiv = (int)C; p.og_hi = S * C;
    const int ch = chunk_elems(X);
    p.a_vec = (No % ch == 0);
    p.x_vec = ((Nd * C) % ch == 0) && (C % ch == 0);
    axis_gemm_launch(p, (int)(B * Se), is_f32(X), stream());
    if (id_skip && !red)  // slot 0 = identity product = X
        fill_slot0(U, X, S);
    return U;
}

// Y[b,m,d,h] = sum_{cs} A2T[(b,)d,cs] V[b,m,cs,h]  (+ bias + optional ReLU).
// V: (B, N, N*S, H) flat (i.e. (B,N,N,S,H) contiguous); A2T: (N, N*S) static
// or (B, N, N*S) dynamic, A2T[d, c*S+s] = Gd[s, c, d]. Out: (B, N, N, H).
// id_skip: A2T is the REDUCED layout A2T'[d, c*(S-1)+s'] = Gd[s'+1, c, d];
// the s = 0 (identity) term Y += V[b, m, (d, 0), h] is added by the epilogue
// (CSUB) while the GEMM contracts only the s >= 1 columns.
torch::Tensor bdgcn_mode2(torch::Tensor V, torch::Tensor A2T,
                          c10::optional<torch::Tensor> bias, bool relu,
                          long N, long S, bool id_skip) {
    check_in(V, "V");
    check_in(A2T, "A2T");
    const bool dyn = A2T.dim() == 3;
    const long B = V.size(0);
    const long Nm = V.size(1);  // origin rows held locally (== N unsharded)
    const long H = V.size(-1);
    const long Se = id_skip ? S - 1 : S;
    TORCH_CHECK(A2T.size(-1) == N * Se && A2T.size(-2) == N, "A2T shape mismatch");
    TORCH_CHECK(B <= 65535, "too many instances");
    auto Y = torch::empty({B, Nm, N, H}, V.options());

    // One GEMM per batch element: OUT[d, (m,h)] = sum_cs A2T[d,cs] V[b,m,cs,h]
    // — folding the m axis into the L (column) dimension gives each staged
    // A-tile (the graph) full-tile MFMA work instead of per-(b,m) instances.
    AxisGemmParams p{};
    p.AT = A2T.data_ptr();
    p.X = V.data_ptr();
    p.OUT = Y.data_ptr();
    p.bias = bias_ptr(bias);
    p.M = (int)N; p.K = (int)(N * Se); p.L = (int)(Nm * H);
    p.a_div = 1; p.a_bs1 = dyn ? N * N * Se : 0; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = Nm * N * S * H; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = Nm * N * H; p.o_bs2 = 0;
    if (id_skip) {
        p.kdiv = (int)Se; p.k_hi = S * H; p.k_lo = H; p.k_base = H;
        p.CSUB = V.data_ptr();
        p.cs_div = 1; p.cs_bs1 = Nm * N * S * H; p.cs_bs2 = 0;
        p.cs_row = S * H;
        p.cs_beta = 1.f;
    } else {
        p.kdiv = 1; p.k_lo = H;
    }
    p.qdiv = (int)H; p.q_hi = N * S * H;   // q = (m, h)
    p.o_row = H;                           // out row = d
    p.ogdiv = (int)H; p.og_hi = N * H;
    p.relu = relu ? 1 : 0;
    p.bias_mod = (int)H;
    const int ch = chunk_elems(V);
    p.a_vec = ((N * Se) % ch == 0);
    p.x_vec = (H % ch == 0);
    axis_gemm_launch(p, (int)B, is_f32(V), stream());
    return Y;
}

// dV[b,m,cs,h] = sum_d A2[(b,)cs,d] dY[b,m,d,h].
// dY: (B, N, N, H); A2: (N*S, N) or (B, N*S, N), A2[c*S+s, d] = Gd[s, c, d].
// Rectangular: dY may be origin-sharded (B, Nm, N, H) with Nm != N
// (region partition, mpgcn_amd/parallel/region.py); N comes from A2.
// id_skip: A2 is the REDUCED layout A2'[c*(S-1)+s', d] = Gd[s'+1, c, d];
// the GEMM writes only the s >= 1 rows of dV (output-row remap) and the
// caller copies dY into the s = 0 rows (dV[..., 0, :] = dY, the identity
// support's exact gradient).
torch::Tensor bdgcn_mode2_bwd(torch::Tensor dY, torch::Tensor A2, long S,
                              bool id_skip, bool no_fill) {
    check_in(dY, "dY");
    check_in(A2, "A2");
    const bool dyn = A2.dim() == 3;
    const long B = dY.size(0), Nm = dY.size(1), H = dY.size(3);
    const long N = A2.size(-1);
    const long Se = id_skip ? S - 1 : S;
    // no_fill (id_skip only): dV carries the Se computed slots without the
    // materialized identity-gradient slot (dY); split-row consumers read dY
    // directly (see bdgcn_mode1 no_fill).
    const bool red = id_skip && no_fill;
    const long So = red ? Se : S;  // slots in dV
    TORCH_CHECK(A2.size(-2) == N * Se && dY.size(2) == N, "A2 shape mismatch");
    TORCH_CHECK(B <= 65535, "too many instances");
    auto dV = torch::empty({B, Nm, N, So, H}, dY.options());

    // One GEMM per batch element (m folded into L, as in bdgcn_mode2):
    // dV[cs, (m,h)] = sum_d A2[cs,d] dY[b,m,d,h]
    AxisGemmParams p{};
    p.AT = A2.data_ptr();
    p.X = dY.data_ptr();
    p.OUT = dV.data_ptr();
    p.M = (int)(N * Se); p.K = (int)N; p.L = (int)(Nm * H);
    p.a_div = 1; p.a_bs1 = dyn ? N * Se * N : 0; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = Nm * N * H; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = Nm * N * So * H; p.o_bs2 = 0;
    p.kdiv = 1; p.k_lo = H;
    p.qdiv = (int)H; p.q_hi = N * H;       // q = (m, h)
    p.o_row = H;                           // out row = cs
    if (id_skip && !red) {
        // GEMM row m = (c, s') lands in dV slot (c, s'+1)
        p.o_mdiv = (int)Se; p.o_m_hi = S * H; p.o_m_lo = H; p.o_m_base = H;
    }
    p.ogdiv = (int)H; p.og_hi = N * So * H;
    const int ch = chunk_elems(dY);
    p.a_vec = (N % ch == 0);
    p.x_vec = (H % ch == 0);
    axis_gemm_launch(p, (int)B, is_f32(dY), stream());
    if (id_skip && !red)  // identity-support gradient rows are dY itself
        fill_slot0(dV, dY, S);
    return dV;
}

// dX[b,n,d,l] = sum_{om} A3T[(b,)n,o*N+m] dU[b,m,d,o,l].
// dU: (B, N, N, S, C); A3T: (N, S*N) or (B, N, S*N), A3T[n, o*N+m] = G[o,n,m].
// Rectangular: dU may be destination-sharded (B, No, Nd, S, C), Nd != No.
// id_skip: A3T is the REDUCED layout A3T'[n, o'*No+m] = Go[o'+1, n, m]; the
// o = 0 (identity) term dX += dU[b, n, d, 0, l] is added by the epilogue.
torch::Tensor bdgcn_mode1_bwd(torch::Tensor dU, torch::Tensor A3T,
                              bool id_skip) {
    check_in(dU, "dU");
    check_in(A3T, "A3T");
    const bool dyn = A3T.dim() == 3;
    const long B = dU.size(0), No = dU.size(1), Nd = dU.size(2);
    const long S = dU.size(3), C = dU.size(4);
    const long Se = id_skip ? S - 1 : S;
    TORCH_CHECK(A3T.size(-2) == No && A3T.size(-1) == Se * No, "A3T shape mismatch");
    TORCH_CHECK(B <= 65535, "too many instances");
    auto dX = torch::empty({B, No, Nd, C}, dU.options());

    AxisGemmParams p{};
    p.AT = A3T.data_ptr();
    p.X = dU.data_ptr();
    p.OUT = dX.data_ptr();
    p.M = (int)No; p.K = (int)(Se * No); p.L = (int)(Nd * C);
    p.a_div = 1; p.a_bs1 = dyn ? No * Se * No : 0; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = No * Nd * S * C; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = No * Nd * C; p.o_bs2 = 0;
    p.kdiv = (int)No; p.k_hi = C; p.k_lo = Nd * S * C;  // k = o*No + m
    if (id_skip) {
        p.k_base = C;  // k = (o', m) addresses support o'+1
        p.CSUB = dU.data_ptr();
        p.cs_div = 1; p.cs_bs1 = No * Nd * S * C; p.cs_bs2 = 0;
        p.cs_row = Nd * S * C;
        p.cs_beta = 1.f;
    }
    p.qdiv = (int)C; p.q_hi = S * C;                    // q = d*C + l
    p.o_row = Nd * C;
    p.ogdiv = 0;
    const int ch = chunk_elems(dU);
    p.a_vec = ((Se * No) % ch == 0);
    p.x_vec = (C % ch == 0);
    axis_gemm_launch(p, (int)B, is_f32(dU), stream());
    return dX;
}

// OUT[r, n] = act(X[r, :] @ W + bias). X: (R, K), W: (K, N), N <= 128.
torch::Tensor row_gemm(torch::Tensor X, torch::Tensor W,
                       c10::optional<torch::Tensor> bias, bool relu) {
    check_in(X, "X");
    check_in(W, "W");
    const long R = X.size(0), K = X.size(1), N = W.size(1);
    TORCH_CHECK(W.size(0) == K, "W shape mismatch");
    TORCH_CHECK(N <= 128, "row_gemm: N must be <= 128");
    TORCH_CHECK(K <= 2048, "row_gemm: K too large for LDS staging");
    auto OUT = torch::empty({R, N}, X.options());
    RowGemmParams p{};
    p.X = X.data_ptr();
    p.W = W.data_ptr();
    p.OUT = OUT.data_ptr();
    p.bias = bias_ptr(bias);
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.o_row = N; p.o_off = 0;
    p.relu = relu ? 1 : 0;
    p.x_vec = (K % chunk_elems(X) == 0);
    row_gemm_launch(p, is_f32(X), stream());
    return OUT;
}

// Same as row_gemm but writes into a caller-provided flat 2-D buffer at
// OUT[r * o_row + o_off + n] — lets Python chunk wide projections (N > 128)
// into column slices of one contiguous output with no extra copy.
void row_gemm_out(torch::Tensor X, torch::Tensor W,
                  c10::optional<torch::Tensor> bias, bool relu,
                  torch::Tensor OUT, long o_row, long o_off) {
    check_in(X, "X");
    check_in(W, "W");
    check_in(OUT, "OUT");
    const long R = X.size(0), K = X.size(1), N = W.size(1);
    TORCH_CHECK(W.size(0) == K, "W shape mismatch");
    TORCH_CHECK(N <= 128, "row_gemm: N must be <= 128");
    TORCH_CHECK(K <= 2048, "row_gemm: K too large for LDS staging");
    RowGemmParams p{};
    p.X = X.data_ptr();
    p.W = W.data_ptr();
    p.OUT = OUT.data_ptr();
    p.bias = bias_ptr(bias);
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.o_row = o_row; p.o_off = o_off;
    p.relu = relu ? 1 : 0;
    p.x_vec = (K % chunk_elems(X) == 0);
    row_gemm_launch(p, is_f32(X), stream());
}

// Split-row row_gemm: logical X row r = [XA[r, :kA] | XB[r, :kB]] — the
// identity-support block is consumed straight from its source tensor (X or
// dY) instead of a materialized slot-0 copy (slot_copy traffic removed).
// Both part widths must be vector-chunk-aligned so fragments never straddle
// the seam (row_gemm.hip split addressing).
torch::Tensor row_gemm_split(torch::Tensor XA, torch::Tensor XB,
                             torch::Tensor W,
                             c10::optional<torch::Tensor> bias, bool relu) {
    check_in(XA, "XA");
    check_in(XB, "XB");
    check_in(W, "W");
    const long R = XA.size(0), kA = XA.size(1), kB = XB.size(1);
    const long K = kA + kB, N = W.size(1);
    TORCH_CHECK(XB.size(0) == R, "row mismatch");
    TORCH_CHECK(XA.scalar_type() == XB.scalar_type(), "dtype mismatch");
    TORCH_CHECK(W.size(0) == K, "W shape mismatch");
    TORCH_CHECK(N <= 128, "row_gemm: N must be <= 128");
    TORCH_CHECK(K <= 2048, "row_gemm: K too large for LDS staging");
    const int ch = chunk_elems(XA);
    TORCH_CHECK(kA % ch == 0 && kB % ch == 0,
                "row_gemm_split: part widths must be chunk-aligned");
    auto OUT = torch::empty({R, N}, XA.options());
    RowGemmParams p{};
    p.X = XB.data_ptr();
    p.X2 = XA.data_ptr();
    p.k0 = (int)kA;
    p.W = W.data_ptr();
    p.OUT = OUT.data_ptr();
    p.bias = bias_ptr(bias);
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.o_row = N; p.o_off = 0;
    p.relu = relu ? 1 : 0;
    p.x_vec = 1;
    row_gemm_launch(p, is_f32(XA), stream());
    return OUT;
}

// Split-row red_gemm for the projection weight gradient: logical rows
// X = [XA | XB] (dY | reduced dV) and Y = [YA | YB] (X | reduced U); both
// identity blocks read from their source tensors, no slot-0 fills.
torch::Tensor red_gemm_split(torch::Tensor XA, torch::Tensor XB,
                             torch::Tensor YA, torch::Tensor YB) {
    check_in(XA, "XA");
    check_in(XB, "XB");
    // fp8 Y parts (fp8-forward mode: YA = X8, YB = reduced U8) route through
    // the YF8 staging with the same split addressing
    const bool y8 = YA.scalar_type() == torch::kFloat8_e4m3fn;
    if (y8) {
        TORCH_CHECK(YB.scalar_type() == torch::kFloat8_e4m3fn &&
                        YA.is_cuda() && YA.is_contiguous() &&
                        YB.is_cuda() && YB.is_contiguous(),
                    "red_gemm_split: fp8 Y parts must both be contiguous fp8");
        TORCH_CHECK(XA.scalar_type() == torch::kBFloat16,
                    "fp8 Y operand requires bf16 X");
    } else {
        check_in(YA, "YA");
        check_in(YB, "YB");
    }
    const long R = XA.size(0);
    const long xkA = XA.size(1), xkB = XB.size(1);
    const long ykA = YA.size(1), ykB = YB.size(1);
    const long K = xkA + xkB, N = ykA + ykB;
    TORCH_CHECK(XB.size(0) == R && YA.size(0) == R && YB.size(0) == R,
                "row mismatch");
    const int ch = chunk_elems(XA);
    TORCH_CHECK(xkA % ch == 0 && xkB % ch == 0 && ykA % ch == 0 &&
                    ykB % ch == 0,
                "red_gemm_split: part widths must be chunk-aligned");
    auto f32 = XA.options().dtype(torch::kFloat);
    const bool det = at::globalContext().deterministicAlgorithms();
    const long nb = det ? red_gemm_nblocks(R) : 1;
    auto out = det ? torch::empty({nb, K, N}, f32) : torch::zeros({K, N}, f32);
    RedGemmParams p{};
    p.X = XB.data_ptr();
    p.x2 = XA.data_ptr();
    p.x_k0 = (int)xkA;
    p.Y = YB.data_ptr();
    p.y2 = YA.data_ptr();
    p.y_k0 = (int)ykA;
    p.out = out.data_ptr<float>();
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.det = det ? 1 : 0;
    p.y_fp8 = y8 ? 1 : 0;
    p.x_vec = 1;
    p.y_vec = 1;
    red_gemm_launch(p, is_f32(XA), stream());
    return det ? out.sum(0) : out;
}

// Reduction GEMM: out = X^T @ Y (f32), plus optional colsum(X) and
// xdot[k] = sum_r X[r,k]*xv[r]. Returns (out, colsum, xdot); colsum/xdot are
// empty tensors when not requested. xv is addressed xv[r*xv_stride + xv_off].
std::vector<torch::Tensor> red_gemm(torch::Tensor X, torch::Tensor Y,
                                    bool want_colsum,
                                    c10::optional<torch::Tensor> xvec,
                                    long xv_stride, long xv_off) {
    check_in(X, "X");
    const bool y8 = Y.scalar_type() == torch::kFloat8_e4m3fn;
    if (y8) {
        TORCH_CHECK(Y.is_cuda() && Y.is_contiguous(), "Y must be contiguous CUDA");
        TORCH_CHECK(X.scalar_type() == torch::kBFloat16,
                    "fp8 Y operand requires bf16 X");
    } else {
        check_in(Y, "Y");
    }
    const long R = X.size(0), K = X.size(1), N = Y.size(1);
    TORCH_CHECK(Y.size(0) == R, "row mismatch");
    auto f32 = X.options().dtype(torch::kFloat);
    const bool has_xv = xvec.has_value() && xvec->defined();
    const bool det = at::globalContext().deterministicAlgorithms();
    const long nb = det ? red_gemm_nblocks(R) : 1;
    // deterministic mode: per-block workspace rows, order-independent sum
    auto out = det ? torch::empty({nb, K, N}, f32) : torch::zeros({K, N}, f32);
    // workspaces are zero-initialized: the kernel's NSPLIT row-partition
    // factor is template-dependent, so some rows may stay unwritten
    auto colsum = !want_colsum ? torch::Tensor()
                  : det ? torch::zeros({2 * nb, K}, f32) : torch::zeros({K}, f32);
    auto xdot = !has_xv ? torch::Tensor()
                : det ? torch::zeros({2 * nb, K}, f32) : torch::zeros({K}, f32);
    RedGemmParams p{};
    p.X = X.data_ptr();
    p.Y = Y.data_ptr();
    p.xvec = has_xv ? xvec->data_ptr() : nullptr;
    p.xv_stride = xv_stride; p.xv_off = xv_off;
    p.out = out.data_ptr<float>();
    p.colsum = want_colsum ? colsum.data_ptr<float>() : nullptr;
    p.xdot = has_xv ? xdot.data_ptr<float>() : nullptr;
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.det = det ? 1 : 0;
    p.y_fp8 = y8 ? 1 : 0;
    const int ch = chunk_elems(X);
    p.x_vec = (K % ch == 0);
    p.y_vec = (N % ch == 0);
    red_gemm_launch(p, is_f32(X), stream());
    if (det) {
        return {out.sum(0), want_colsum ? colsum.sum(0) : colsum,
                has_xv ? xdot.sum(0) : xdot};
    }
    return {out, colsum, xdot};
}

// One fused LSTM step, writing into caller-provided output slices (lets the
// Python layer keep per-timestep states in contiguous T-slab buffers so the
// whole sequence's weight-grad reduction is a single red_gemm pass).
// x[r * x_stride + x_off]; h_prev/h_out: (R, H); c_prev/c_out: (R, H) f32;
// whh: (4H, H); wih, bias: (4H) f32; gates_out: (R, 4H) post-activation.
void lstm_step_fwd(torch::Tensor x, long x_stride, long x_off,
                   torch::Tensor h_prev, torch::Tensor c_prev,
                   torch::Tensor whh, torch::Tensor wih, torch::Tensor bias,
                   torch::Tensor h_out, torch::Tensor c_out,
                   torch::Tensor gates_out) {
    check_in(x, "x");
    check_in(h_prev, "h_prev");
    check_in(whh, "whh");
    check_in(h_out, "h_out");
    check_in(gates_out, "gates_out");
    TORCH_CHECK(c_prev.scalar_type() == torch::kFloat && c_prev.is_contiguous());
    TORCH_CHECK(c_out.scalar_type() == torch::kFloat && c_out.is_contiguous());
    TORCH_CHECK(wih.scalar_type() == torch::kFloat && bias.scalar_type() == torch::kFloat);
    const long R = h_prev.size(0), H = h_prev.size(1);
    LstmStepParams p{};
    p.x = x.data_ptr(); p.x_stride = x_stride; p.x_off = x_off;
    p.h_prev = h_prev.data_ptr();
    p.c_prev = c_prev.data_ptr<float>();
    p.whh = whh.data_ptr();
    p.wih = wih.data_ptr<float>();
    p.bias = bias.data_ptr<float>();
    p.h_out = h_out.data_ptr();
    p.c_out = c_out.data_ptr<float>();
    p.gates_out = gates_out.data_ptr();
    p.R = R; p.H = (int)H;
    lstm_step_fwd_launch(p, is_f32(h_prev), stream());
}

// One LSTM backward step (pointwise part), writing into provided buffers.
void lstm_step_bwd(torch::Tensor dh, c10::optional<torch::Tensor> dc_in,
                   torch::Tensor gates, torch::Tensor c_prev, torch::Tensor c,
                   torch::Tensor dgates_out, torch::Tensor dc_prev_out) {
    check_in(dh, "dh");
    check_in(gates, "gates");
    check_in(dgates_out, "dgates_out");
    const long R = dh.size(0), H = dh.size(1);
    LstmBwdParams p{};
    p.dh = dh.data_ptr();
    p.dc_in = (dc_in.has_value() && dc_in->defined()) ? dc_in->data_ptr<float>() : nullptr;
    p.gates = gates.data_ptr();
    p.c_prev = c_prev.data_ptr<float>();
    p.c = c.data_ptr<float>();
    p.dgates = dgates_out.data_ptr();
    p.dc_prev = dc_prev_out.data_ptr<float>();
    p.R = R; p.H = (int)H;
    lstm_step_bwd_launch(p, is_f32(dh), stream());
}

// Fully-fused register-resident LSTM forward over ONE chunk of <= 8 steps:
// x (R, x_cols) bf16, columns [x_off, x_off+T). Chunks longer sequences
// chain through (h_in, c_in) boundary checkpoints (ops/functional.py) —
// the register-resident schedule then covers ANY T with O(R*H) checkpoint
// traffic instead of the slab path's O(R*H*T) state round trips.
// Returns {h_end (R,32) bf16, c_end (R,32) f32 or undefined}.
std::vector<torch::Tensor> lstm_fused_fwd(torch::Tensor x, long x_off,
                                          long T_logical, torch::Tensor whh,
                                          torch::Tensor wih, torch::Tensor bias,
                                          c10::optional<torch::Tensor> h_in,
                                          c10::optional<torch::Tensor> c_in,
                                          bool want_c) {
    check_in(x, "x");
    check_in(whh, "whh");
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "fused LSTM is bf16-only");
    TORCH_CHECK(whh.size(1) == 32 && whh.size(0) == 128, "fused LSTM needs H=32");
    TORCH_CHECK(x.size(1) % 8 == 0 && x_off % 8 == 0 && x_off + 8 <= x.size(1),
                "x must be zero-padded to a multiple of 8 columns");
    const long R = x.size(0);
    const int T = (int)T_logical;
    TORCH_CHECK(T >= 1 && T <= 8, "fused LSTM chunk needs T <= 8");
    const bool has_h = h_in.has_value() && h_in->defined();
    auto h = torch::empty({R, 32}, x.options());
    auto c = want_c ? torch::empty({R, 32}, x.options().dtype(torch::kFloat))
                    : torch::Tensor();
    LstmFusedParams p{};
    p.x = x.data_ptr();
    p.x_cols = x.size(1); p.x_off = x_off;
    p.whh = whh.data_ptr();
    p.wih = wih.data_ptr<float>();
    p.bias = bias.data_ptr<float>();
    p.h_out = h.data_ptr();
    p.h_in = has_h ? h_in->data_ptr() : nullptr;
    p.c_in = has_h ? c_in->data_ptr<float>() : nullptr;
    p.c_out = want_c ? c.data_ptr<float>() : nullptr;
    p.R = R; p.T = T;
    lstm_fused_fwd_launch(p, stream());
    return {h, c};
}

// Fused backward of one chunk with in-kernel forward recompute from the
// chunk-entry checkpoint (h_in, c_in). dc_in chains dL/dc from the following
// chunk; want_prev emits {dh_prev, dc_prev} = gradients at the chunk entry
// for the preceding chunk's backward. dx (full (R, x_cols) buffer) gets this
// chunk's columns written in place. Returns
// {dwhh (4H,H) f32, dbias (4H) f32, dwih (4H) f32, dh_prev, dc_prev}.
std::vector<torch::Tensor> lstm_fused_bwd(torch::Tensor x, long x_off,
                                          long T_logical,
                                          torch::Tensor whh,
                                          torch::Tensor whh2, torch::Tensor wih,
                                          torch::Tensor bias, torch::Tensor dh,
                                          c10::optional<torch::Tensor> h_in,
                                          c10::optional<torch::Tensor> c_in,
                                          c10::optional<torch::Tensor> dc_in,
                                          bool want_prev,
                                          c10::optional<torch::Tensor> dx) {
    check_in(x, "x");
    check_in(whh, "whh");
    check_in(whh2, "whh2");
    check_in(dh, "dh");
    TORCH_CHECK(x.size(1) % 8 == 0 && x_off % 8 == 0 && x_off + 8 <= x.size(1),
                "x must be zero-padded to a multiple of 8 columns");
    const long R = x.size(0);
    const int T = (int)T_logical;
    const int nb = lstm_fused_bwd_blocks(R);
    const bool has_h = h_in.has_value() && h_in->defined();
    const bool has_dc = dc_in.has_value() && dc_in->defined();
    const bool has_dx = dx.has_value() && dx->defined();
    auto f32 = x.options().dtype(torch::kFloat);
    auto ws_dw = torch::empty({nb, 128, 32}, f32);
    auto ws_db = torch::empty({nb, 128}, f32);
    auto ws_dwih = torch::empty({nb, 128}, f32);
    auto dh_prev = want_prev ? torch::empty({R, 32}, x.options()) : torch::Tensor();
    auto dc_prev = want_prev ? torch::empty({R, 32}, f32) : torch::Tensor();
    LstmFusedParams p{};
    p.x = x.data_ptr();
    p.x_cols = x.size(1); p.x_off = x_off;
    p.whh = whh.data_ptr();
    p.whh2 = whh2.data_ptr();
    p.wih = wih.data_ptr<float>();
    p.bias = bias.data_ptr<float>();
    p.dh = dh.data_ptr();
    p.h_in = has_h ? h_in->data_ptr() : nullptr;
    p.c_in = has_h ? c_in->data_ptr<float>() : nullptr;
    p.dc_in = has_dc ? dc_in->data_ptr<float>() : nullptr;
    p.dh_out = want_prev ? dh_prev.data_ptr() : nullptr;
    p.dc_out = want_prev ? dc_prev.data_ptr<float>() : nullptr;
    p.ws_dw = ws_dw.data_ptr<float>();
    p.ws_db = ws_db.data_ptr<float>();
    p.ws_dwih = ws_dwih.data_ptr<float>();
    p.dx = has_dx ? dx->data_ptr() : nullptr;
    p.R = R; p.T = T;
    lstm_fused_bwd_launch(p, stream());
    // in-kernel fixed-order workspace reduction (deterministic; three
    // aten::sum launches were wall-expensive under branch-stream overlap)
    auto dwhh = torch::empty({128, 32}, f32);
    auto dbias = torch::empty({128}, f32);
    auto dwih = torch::empty({128}, f32);
    slab_colsum3_launch(ws_dw.data_ptr<float>(), dwhh.data_ptr<float>(), 128 * 32,
                        ws_db.data_ptr<float>(), dbias.data_ptr<float>(), 128,
                        ws_dwih.data_ptr<float>(), dwih.data_ptr<float>(), 128,
                        nb, stream());
    return {dwhh, dbias, dwih, dh_prev, dc_prev};
}

// fp8 probe of the mode-2 contraction (docs/ROADMAP.md byte-reduction
// lever; measurement-only — the training path stays bf16). V8/A2T8 are
// torch float8_e4m3fn (bit-compatible with gfx950's OCP e4m3); returns a
// float8_e4m3fn Y of the usual (B, Nm, N, H) mode-2 shape.
torch::Tensor bdgcn_mode2_fp8(torch::Tensor V8, torch::Tensor A2T8,
                              c10::optional<torch::Tensor> bias, bool relu,
                              long N, long S) {
    TORCH_CHECK(V8.is_cuda() && V8.is_contiguous() &&
                V8.scalar_type() == torch::kFloat8_e4m3fn, "V8 must be fp8");
    TORCH_CHECK(A2T8.is_cuda() && A2T8.is_contiguous() &&
                A2T8.scalar_type() == torch::kFloat8_e4m3fn, "A2T8 must be fp8");
    const long B = V8.size(0), Nm = V8.size(1), H = V8.size(-1);
    TORCH_CHECK(A2T8.size(-1) == N * S && A2T8.size(-2) == N, "A2T8 shape");
    TORCH_CHECK(N % 256 == 0 && (N * S) % 128 == 0 && (Nm * H) % 256 == 0,
                "fp8 probe requires full tiles");
    auto Y = torch::empty({B, Nm, N, H}, V8.options());

    AxisGemmParams p{};
    p.AT = A2T8.data_ptr();
    p.X = V8.data_ptr();
    p.OUT = Y.data_ptr();
    p.bias = bias_ptr(bias);
    p.M = (int)N; p.K = (int)(N * S); p.L = (int)(Nm * H);
    p.a_div = 1; p.a_bs1 = 0; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = Nm * N * S * H; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = Nm * N * H; p.o_bs2 = 0;
    p.kdiv = 1; p.k_lo = H;
    p.qdiv = (int)H; p.q_hi = N * S * H;
    p.o_row = H;
    p.ogdiv = (int)H; p.og_hi = N * H;
    p.relu = relu ? 1 : 0;
    p.bias_mod = (int)H;
    p.a_vec = ((N * S) % 16 == 0);
    p.x_vec = (H % 16 == 0);
    axis_gemm_fp8_launch(p, (int)B, 0, stream());
    return Y;
}

// fp8 probe of the mode-1 contraction (same measurement-only contract as
// bdgcn_mode2_fp8; square static graphs, full tiles).
torch::Tensor bdgcn_mode1_fp8(torch::Tensor X8, torch::Tensor GT8) {
    TORCH_CHECK(X8.is_cuda() && X8.is_contiguous() &&
                X8.scalar_type() == torch::kFloat8_e4m3fn, "X8 must be fp8");
    TORCH_CHECK(GT8.is_cuda() && GT8.is_contiguous() &&
                GT8.scalar_type() == torch::kFloat8_e4m3fn, "GT8 must be fp8");
    const long B = X8.size(0), N = X8.size(1), C = X8.size(3);
    const long S = GT8.size(0);
    TORCH_CHECK(GT8.dim() == 3 && GT8.size(1) == N && GT8.size(2) == N, "GT8 shape");
    TORCH_CHECK(N % 128 == 0 && (N * C) % 256 == 0, "fp8 probe requires full tiles");
    auto U = torch::empty({B, N, N, S, C}, X8.options());
    AxisGemmParams p{};
    p.AT = GT8.data_ptr();
    p.X = X8.data_ptr();
    p.OUT = U.data_ptr();
    p.M = (int)N; p.K = (int)N; p.L = (int)(N * C);
    p.a_div = (int)S; p.a_bs1 = 0; p.a_bs2 = N * N;
    p.x_div = (int)S; p.x_bs1 = N * N * C; p.x_bs2 = 0;
    p.o_div = (int)S; p.o_bs1 = N * N * S * C; p.o_bs2 = C;
    p.kdiv = 1; p.k_lo = N * C;
    p.qdiv = 0;
    p.o_row = N * S * C;
    p.ogdiv = (int)C; p.og_hi = S * C;
    p.a_vec = (N % 16 == 0);
    p.x_vec = ((N * C) % 16 == 0) && (C % 16 == 0);
    axis_gemm_fp8_launch(p, (int)(B * S), 0, stream());
    return U;
}

namespace {
void check_fp8(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                t.scalar_type() == torch::kFloat8_e4m3fn,
                name, " must be contiguous CUDA float8_e4m3fn");
}
}  // namespace

// fp8-forward training mode-1: U8 = mode1(X8, GT8), fp8 OUTPUT ONLY — U8
// both feeds the fp8 projection GEMM and is the saved backward operand (the
// dW reduction red_gemm reads fp8 Y directly), so mode-1's output traffic is
// HALF the bf16 path's (U is the largest tensor in the step) and no bf16
// twin is written. Forward math runs entirely on fp8 operands (half the
// staged bytes through the byte-bound load path — the measured 1.30x kernel
// lever); the gradient CONTRACTIONS stay bf16. Shape contract: the fp8 tile
// is vector-only, so the column extent Nd*C must be a multiple of 256 and C
// a multiple of 16 (flagship and large-N configs satisfy this; the Python
// layer falls back to bf16 otherwise).
torch::Tensor bdgcn_mode1_fp8_train(torch::Tensor X8, torch::Tensor GT8,
                                    bool id_skip, bool no_fill) {
    check_fp8(X8, "X8");
    check_fp8(GT8, "GT8");
    const bool dyn = GT8.dim() == 4;
    const long B = X8.size(0), No = X8.size(1), Nd = X8.size(2), C = X8.size(3);
    const long Se = dyn ? GT8.size(1) : GT8.size(0);
    // no_fill: identity-slot-free U8 (see bdgcn_mode1) — consumers read the
    // identity block straight from X8 via the split-row fp8 kernels
    const bool red = id_skip && no_fill;
    const long S = (id_skip && !red) ? Se + 1 : Se;
    TORCH_CHECK(GT8.size(-1) == No && GT8.size(-2) == No, "shape mismatch");
    TORCH_CHECK(!dyn || GT8.size(0) == B, "dynamic GT8 batch mismatch");
    TORCH_CHECK(B * Se <= 65535, "too many instances");
    TORCH_CHECK((Nd * C) % 256 == 0 && C % 16 == 0,
                "fp8 mode-1 needs (Nd*C) % 256 == 0 and C % 16 == 0");
    auto U8 = torch::empty({B, No, Nd, S, C}, X8.options());

    AxisGemmParams p{};
    p.AT = GT8.data_ptr();
    p.X = X8.data_ptr();
    p.OUT = (id_skip && !red) ? (void*)((char*)U8.data_ptr() + C)
                              : U8.data_ptr();
    p.M = (int)No; p.K = (int)No; p.L = (int)(Nd * C);
    p.a_div = (int)Se; p.a_bs1 = dyn ? Se * No * No : 0; p.a_bs2 = No * No;
    p.x_div = (int)Se; p.x_bs1 = No * Nd * C; p.x_bs2 = 0;
    p.o_div = (int)Se; p.o_bs1 = No * Nd * S * C; p.o_bs2 = C;
    p.kdiv = 1; p.k_lo = Nd * C;
    p.qdiv = 0;
    p.o_row = Nd * S * C;
    p.ogdiv = (int)C; p.og_hi = S * C;
    p.a_vec = (No % 16 == 0);
    p.x_vec = 1;
    axis_gemm_fp8_launch(p, (int)(B * Se), 0, stream());
    if (id_skip && !red)
        fill_slot0(U8, X8, S);
    return U8;
}

// fp8-forward training mode-2: Y = mode2(V8, A2T8) + bias + act with DUAL
// outputs — Y_bf16 (the autograd output / ReLU mask operand) and Y8 (fp8
// twin that feeds the NEXT layer's mode-1 without a separate quantize pass).
std::vector<torch::Tensor> bdgcn_mode2_fp8_train(torch::Tensor V8,
                                                 torch::Tensor A2T8,
                                                 c10::optional<torch::Tensor> bias,
                                                 bool relu, long N, long S,
                                                 bool want_twin, bool id_skip) {
    check_fp8(V8, "V8");
    check_fp8(A2T8, "A2T8");
    const bool dyn = A2T8.dim() == 3;
    const long B = V8.size(0), Nm = V8.size(1), H = V8.size(-1);
    const long Se = id_skip ? S - 1 : S;
    TORCH_CHECK(A2T8.size(-1) == N * Se && A2T8.size(-2) == N, "A2T8 shape");
    TORCH_CHECK(B <= 65535, "too many instances");
    TORCH_CHECK((Nm * H) % 256 == 0 && H % 16 == 0,
                "fp8 mode-2 needs (Nm*H) % 256 == 0 and H % 16 == 0");
    auto Y = torch::empty({B, Nm, N, H}, V8.options().dtype(torch::kBFloat16));
    // want_twin=false (e.g. the LAST gcn layer, whose output feeds the bf16
    // FC head) skips the fp8 twin write entirely
    auto Y8 = want_twin ? torch::empty({B, Nm, N, H}, V8.options())
                        : torch::Tensor();

    AxisGemmParams p{};
    p.AT = A2T8.data_ptr();
    p.X = V8.data_ptr();
    p.OUT = Y.data_ptr();
    p.OUT2 = want_twin ? Y8.data_ptr() : nullptr;
    p.bias = bias_ptr(bias);
    p.M = (int)N; p.K = (int)(N * Se); p.L = (int)(Nm * H);
    p.a_div = 1; p.a_bs1 = dyn ? N * N * Se : 0; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = Nm * N * S * H; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = Nm * N * H; p.o_bs2 = 0;
    if (id_skip) {
        p.kdiv = (int)Se; p.k_hi = S * H; p.k_lo = H; p.k_base = H;
        p.CSUB = V8.data_ptr();
        p.cs_div = 1; p.cs_bs1 = Nm * N * S * H; p.cs_bs2 = 0;
        p.cs_row = S * H;
        p.cs_beta = 1.f;
    } else {
        p.kdiv = 1; p.k_lo = H;
    }
    p.qdiv = (int)H; p.q_hi = N * S * H;
    p.o_row = H;
    p.ogdiv = (int)H; p.og_hi = N * H;
    p.relu = relu ? 1 : 0;
    p.bias_mod = (int)H;
    p.a_vec = ((N * Se) % 16 == 0);
    p.x_vec = 1;
    axis_gemm_fp8_launch(p, (int)B, 2, stream());
    return {Y, Y8};
}

// fp8 gradient contraction dV = mode2_bwd(dY8, A28), SCALED: dY was
// quantized as dY8 = fp8(dY * s) with a device-resident dynamic scale
// (s = margin/amax(dY) — gradients underflow e4m3's 2^-9 floor without it);
// the epilogue multiplies by *inv_scale (device pointer, no host sync) and
// writes bf16 dV. A28[cs, d] = fp8(Gd[s, c, d]) — supports are O(1), unit
// scale. Same contraction as bdgcn_mode2_bwd (bf16 twin of this path).
torch::Tensor bdgcn_mode2_bwd_fp8(torch::Tensor dY8, torch::Tensor A28,
                                  long S, torch::Tensor inv_scale,
                                  torch::Tensor dY_bf16, bool id_skip,
                                  bool no_fill) {
    check_fp8(dY8, "dY8");
    check_fp8(A28, "A28");
    TORCH_CHECK(inv_scale.is_cuda() && inv_scale.scalar_type() == torch::kFloat,
                "inv_scale must be a CUDA f32 scalar tensor");
    const bool dyn = A28.dim() == 3;
    const long B = dY8.size(0), Nm = dY8.size(1), H = dY8.size(3);
    const long N = A28.size(-1);
    const long Se = id_skip ? S - 1 : S;
    // no_fill: reduced dV, no materialized identity-gradient slot — the
    // split-row consumers read the exact bf16 dY directly (same numerics)
    const bool red = id_skip && no_fill;
    const long So = red ? Se : S;
    TORCH_CHECK(A28.size(-2) == N * Se && dY8.size(2) == N, "A28 shape");
    TORCH_CHECK(B <= 65535, "too many instances");
    TORCH_CHECK((Nm * H) % 256 == 0 && H % 16 == 0, "fp8 bwd shape gate");
    auto dV = torch::empty({B, Nm, N, So, H},
                           dY8.options().dtype(torch::kBFloat16));
    AxisGemmParams p{};
    p.AT = A28.data_ptr();
    p.X = dY8.data_ptr();
    p.OUT = dV.data_ptr();
    p.scale = inv_scale.data_ptr<float>();
    p.M = (int)(N * Se); p.K = (int)N; p.L = (int)(Nm * H);
    p.a_div = 1; p.a_bs1 = dyn ? N * Se * N : 0; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = Nm * N * H; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = Nm * N * So * H; p.o_bs2 = 0;
    p.kdiv = 1; p.k_lo = H;
    p.qdiv = (int)H; p.q_hi = N * H;
    p.o_row = H;
    if (id_skip && !red) {
        p.o_mdiv = (int)Se; p.o_m_hi = S * H; p.o_m_lo = H; p.o_m_base = H;
    }
    p.ogdiv = (int)H; p.og_hi = N * So * H;
    p.a_vec = (N % 16 == 0);
    p.x_vec = 1;
    axis_gemm_fp8_launch(p, (int)B, 2, stream());
    if (id_skip && !red)  // identity-support gradient rows: exact bf16 dY
        fill_slot0(dV, dY_bf16, S);
    return dV;
}

// fp8 gradient contraction dX = mode1_bwd(dU8, A3T8), scaled like
// bdgcn_mode2_bwd_fp8. dU8 = fp8(dU * s); epilogue descales and writes bf16.
torch::Tensor bdgcn_mode1_bwd_fp8(torch::Tensor dU8, torch::Tensor A3T8,
                                  torch::Tensor inv_scale, bool id_skip) {
    check_fp8(dU8, "dU8");
    check_fp8(A3T8, "A3T8");
    TORCH_CHECK(inv_scale.is_cuda() && inv_scale.scalar_type() == torch::kFloat,
                "inv_scale must be a CUDA f32 scalar tensor");
    const bool dyn = A3T8.dim() == 3;
    const long B = dU8.size(0), No = dU8.size(1), Nd = dU8.size(2);
    const long S = dU8.size(3), C = dU8.size(4);
    const long Se = id_skip ? S - 1 : S;
    TORCH_CHECK(A3T8.size(-2) == No && A3T8.size(-1) == Se * No, "A3T8 shape");
    TORCH_CHECK(B <= 65535, "too many instances");
    TORCH_CHECK((Nd * C) % 256 == 0 && C % 16 == 0, "fp8 bwd shape gate");
    auto dX = torch::empty({B, No, Nd, C}, dU8.options().dtype(torch::kBFloat16));
    AxisGemmParams p{};
    p.AT = A3T8.data_ptr();
    p.X = dU8.data_ptr();
    p.OUT = dX.data_ptr();
    p.scale = inv_scale.data_ptr<float>();
    p.M = (int)No; p.K = (int)(Se * No); p.L = (int)(Nd * C);
    p.a_div = 1; p.a_bs1 = dyn ? No * Se * No : 0; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = No * Nd * S * C; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = No * Nd * C; p.o_bs2 = 0;
    p.kdiv = (int)No; p.k_hi = C; p.k_lo = Nd * S * C;  // k = o*No + m
    if (id_skip) {
        p.k_base = C;
        p.CSUB = dU8.data_ptr();
        p.cs_div = 1; p.cs_bs1 = No * Nd * S * C; p.cs_bs2 = 0;
        p.cs_row = Nd * S * C;
        p.cs_beta = 1.f;
    }
    p.qdiv = (int)C; p.q_hi = S * C;                    // q = d*C + l
    p.o_row = Nd * C;
    p.ogdiv = 0;
    p.a_vec = ((Se * No) % 16 == 0);
    p.x_vec = 1;
    axis_gemm_fp8_launch(p, (int)B, 2, stream());
    return dX;
}

// fp8 projection GEMM: OUT8[r, n] = X8[r, :] @ W8 — the fp8-forward mode's
// V = U @ Wre (output feeds fp8 mode-2; V is not needed by backward, so no
// bf16 twin). K must be a multiple of 16 (vectorized fp8 row reads).
torch::Tensor row_gemm_fp8(torch::Tensor X8, torch::Tensor W8) {
    check_fp8(X8, "X8");
    check_fp8(W8, "W8");
    const long R = X8.size(0), K = X8.size(1), N = W8.size(1);
    TORCH_CHECK(W8.size(0) == K, "W8 shape mismatch");
    TORCH_CHECK(N <= 128, "row_gemm_fp8: N must be <= 128");
    TORCH_CHECK(K % 16 == 0 && K <= 2048, "row_gemm_fp8: K % 16 != 0 or too large");
    auto OUT = torch::empty({R, N}, X8.options());
    RowGemmParams p{};
    p.X = X8.data_ptr();
    p.W = W8.data_ptr();
    p.OUT = OUT.data_ptr();
    p.bias = nullptr;
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.o_row = N; p.o_off = 0;
    p.relu = 0;
    p.x_vec = 1;
    row_gemm_fp8_launch(p, stream());
    return OUT;
}

// Split-row fp8 projection: logical rows [XA8 | XB8] — the identity block
// (X8) streams from its source, no fp8 slot-0 fill (see row_gemm_split).
torch::Tensor row_gemm_fp8_split(torch::Tensor XA8, torch::Tensor XB8,
                                 torch::Tensor W8) {
    check_fp8(XA8, "XA8");
    check_fp8(XB8, "XB8");
    check_fp8(W8, "W8");
    const long R = XA8.size(0), kA = XA8.size(1), kB = XB8.size(1);
    const long K = kA + kB, N = W8.size(1);
    TORCH_CHECK(XB8.size(0) == R, "row mismatch");
    TORCH_CHECK(W8.size(0) == K, "W8 shape mismatch");
    TORCH_CHECK(N <= 128, "row_gemm_fp8: N must be <= 128");
    TORCH_CHECK(kA % 16 == 0 && kB % 16 == 0 && K <= 2048,
                "row_gemm_fp8_split: part widths must be 16-aligned");
    auto OUT = torch::empty({R, N}, XA8.options());
    RowGemmParams p{};
    p.X = XB8.data_ptr();
    p.X2 = XA8.data_ptr();
    p.k0 = (int)kA;
    p.W = W8.data_ptr();
    p.OUT = OUT.data_ptr();
    p.bias = nullptr;
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.o_row = N; p.o_off = 0;
    p.relu = 0;
    p.x_vec = 1;
    row_gemm_fp8_launch(p, stream());
    return OUT;
}

// relu_bwd_colsum with a FUSED fp8 gradient output: dY8 = fp8(dY * *q_scale)
// written in the same streaming pass; this step's amax(|dY|) is recorded
// into amax_out for the next step's delayed scale (ops/functional.py).
std::vector<torch::Tensor> relu_bwd_colsum_fp8(torch::Tensor dH, torch::Tensor Y,
                                               bool mask, torch::Tensor q_scale,
                                               torch::Tensor amax_out) {
    check_in(dH, "dH");
    TORCH_CHECK(dH.scalar_type() == torch::kBFloat16, "bf16 only");
    TORCH_CHECK(q_scale.is_cuda() && q_scale.scalar_type() == torch::kFloat);
    TORCH_CHECK(amax_out.is_cuda() && amax_out.scalar_type() == torch::kFloat);
    const long H = dH.size(-1);
    TORCH_CHECK(H >= 8 && (H & (H - 1)) == 0 && H <= 2048, "bad H");
    auto dY = torch::empty_like(dH);
    auto dY8 = torch::empty(dH.sizes(), dH.options().dtype(torch::kFloat8_e4m3fn));
    const bool det = at::globalContext().deterministicAlgorithms();
    const long nb = det ? relu_bwd_nblocks(dH.numel()) : 1;
    auto colsum = det ? torch::empty({nb, H}, dH.options().dtype(torch::kFloat))
                      : torch::zeros({H}, dH.options().dtype(torch::kFloat));
    ReluBwdParams p{};
    p.dH = dH.data_ptr();
    p.Y = mask ? Y.data_ptr() : dH.data_ptr();
    p.dY = dY.data_ptr();
    p.dY8 = dY8.data_ptr();
    p.q_scale = q_scale.data_ptr<float>();
    p.amax_out = amax_out.data_ptr<float>();
    p.colsum = colsum.data_ptr<float>();
    p.total = dH.numel();
    p.H = (int)H;
    p.mask = mask ? 1 : 0;
    p.det = det ? 1 : 0;
    relu_bwd_colsum_launch(p, stream());
    if (det) {
        auto cs = torch::empty({H}, dH.options().dtype(torch::kFloat));
        slab_colsum_launch(colsum.data_ptr<float>(), cs.data_ptr<float>(), nb,
                           H, stream());
        return {dY, dY8, cs};
    }
    return {dY, dY8, colsum};
}

// row_gemm emitting a SCALED fp8 output only (no bf16 store): the fp8-mode
// dU projection — its sole consumer is the fp8 dX contraction, so the bf16
// copy is never materialized. Tracks amax(|v|) into amax_out for the next
// step's delayed scale.
torch::Tensor row_gemm_fp8_out(torch::Tensor X, torch::Tensor W,
                               torch::Tensor q_scale, torch::Tensor amax_out) {
    check_in(X, "X");
    check_in(W, "W");
    TORCH_CHECK(X.scalar_type() == torch::kBFloat16, "bf16 only");
    TORCH_CHECK(q_scale.is_cuda() && q_scale.scalar_type() == torch::kFloat);
    TORCH_CHECK(amax_out.is_cuda() && amax_out.scalar_type() == torch::kFloat);
    const long R = X.size(0), K = X.size(1), N = W.size(1);
    TORCH_CHECK(W.size(0) == K, "W shape mismatch");
    TORCH_CHECK(N <= 128 && K <= 2048, "row_gemm shape gate");
    auto OUT8 = torch::empty({R, N}, X.options().dtype(torch::kFloat8_e4m3fn));
    RowGemmParams p{};
    p.X = X.data_ptr();
    p.W = W.data_ptr();
    p.OUT = nullptr;
    p.OUT8 = OUT8.data_ptr();
    p.q_scale = q_scale.data_ptr<float>();
    p.amax_out = amax_out.data_ptr<float>();
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.o_row = N; p.o_off = 0;
    p.relu = 0;
    p.x_vec = (K % chunk_elems(X) == 0);
    row_gemm_launch(p, 0, stream());
    return OUT8;
}

// Split-row variant of row_gemm_fp8_out: logical bf16 rows [XA | XB]
// (dY | reduced dV) -> scaled fp8 dU8, identity-gradient block read from dY.
torch::Tensor row_gemm_fp8_out_split(torch::Tensor XA, torch::Tensor XB,
                                     torch::Tensor W, torch::Tensor q_scale,
                                     torch::Tensor amax_out) {
    check_in(XA, "XA");
    check_in(XB, "XB");
    check_in(W, "W");
    TORCH_CHECK(XA.scalar_type() == torch::kBFloat16 &&
                    XB.scalar_type() == torch::kBFloat16, "bf16 only");
    TORCH_CHECK(q_scale.is_cuda() && q_scale.scalar_type() == torch::kFloat);
    TORCH_CHECK(amax_out.is_cuda() && amax_out.scalar_type() == torch::kFloat);
    const long R = XA.size(0), kA = XA.size(1), kB = XB.size(1);
    const long K = kA + kB, N = W.size(1);
    TORCH_CHECK(XB.size(0) == R, "row mismatch");
    TORCH_CHECK(W.size(0) == K, "W shape mismatch");
    TORCH_CHECK(N <= 128 && K <= 2048, "row_gemm shape gate");
    const int ch = chunk_elems(XA);
    TORCH_CHECK(kA % ch == 0 && kB % ch == 0,
                "row_gemm_fp8_out_split: part widths must be chunk-aligned");
    auto OUT8 = torch::empty({R, N}, XA.options().dtype(torch::kFloat8_e4m3fn));
    RowGemmParams p{};
    p.X = XB.data_ptr();
    p.X2 = XA.data_ptr();
    p.k0 = (int)kA;
    p.W = W.data_ptr();
    p.OUT = nullptr;
    p.OUT8 = OUT8.data_ptr();
    p.q_scale = q_scale.data_ptr<float>();
    p.amax_out = amax_out.data_ptr<float>();
    p.R = R; p.K = (int)K; p.N = (int)N;
    p.o_row = N; p.o_off = 0;
    p.relu = 0;
    p.x_vec = 1;
    row_gemm_launch(p, 0, stream());
    return OUT8;
}

// Device-side delayed-scaling step: scale = margin/max(amax, eps);
// inv = 1/scale; amax = 0. No host synchronization.
void fp8_scale_update(torch::Tensor amax, torch::Tensor scale,
                      torch::Tensor inv, double margin) {
    fp8_scale_update_launch(amax.data_ptr<float>(), scale.data_ptr<float>(),
                            inv.data_ptr<float>(), (float)margin, stream());
}

// Fused flat-buffer Adam step (one launch + a 1-thread step bump). All
// operands are packed f32 device buffers; the step counter t is a 1-element
// f32 tensor that lives on device, so the call is hipGraph-capture-safe.
void adam_flat(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor t, double lr, double b1,
               double b2, double eps, double wd) {
    TORCH_CHECK(p.is_cuda() && p.is_contiguous() &&
                    p.scalar_type() == torch::kFloat32,
                "adam_flat: p must be contiguous f32 CUDA");
    TORCH_CHECK(g.numel() == p.numel() && m.numel() == p.numel() &&
                    v.numel() == p.numel(),
                "adam_flat: buffer size mismatch");
    adam_flat_launch(p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     t.data_ptr<float>(), (long)p.numel(), (float)lr, (float)b1,
                     (float)b2, (float)eps, (float)wd, stream());
}

// Fused dynamic-support build (K8, random_walk_diffusion): raw flow
// (B, N, N) f32 -> (B, order+1, N, N) f32 support stack in 2 + (order-1)
// launches (rowsum, fused normalize+transpose+identity, then one
// alpha/CSUB-epilogue GEMM per Chebyshev order). Replaces the per-step
// ~10-launch stock chain (graph/supports.py torch path, kept as the
// CPU/other-kernel-type fallback and the numerics oracle).
torch::Tensor rwd_supports(torch::Tensor flow, long order) {
    TORCH_CHECK(flow.is_cuda() && flow.is_contiguous() &&
                flow.scalar_type() == torch::kFloat, "flow must be CUDA f32");
    TORCH_CHECK(flow.dim() == 3 && flow.size(1) == flow.size(2), "flow (B,N,N)");
    TORCH_CHECK(order >= 1, "rwd_supports needs order >= 1");
    const long B = flow.size(0), N = flow.size(1), S = order + 1;
    TORCH_CHECK(B <= 65535, "too many instances");
    auto out = torch::empty({B, S, N, N}, flow.options());
    auto d = torch::empty({B, N}, flow.options());
    rwd_rowsum_launch(flow.data_ptr<float>(), d.data_ptr<float>(), B * N,
                      (int)N, stream());
    rwd_norm_t_launch(flow.data_ptr<float>(), d.data_ptr<float>(),
                      out.data_ptr<float>(), B, (int)N, S * N * N, stream());
    float* base = out.data_ptr<float>();
    for (long k = 2; k <= order; ++k) {
        // T_k = 2 * PT @ T_{k-1} - T_{k-2}
        AxisGemmParams p{};
        p.AT = base + 1 * N * N;        // PT (slot 1)
        p.X = base + (k - 1) * N * N;   // T_{k-1}
        p.OUT = base + k * N * N;       // T_k
        p.CSUB = base + (k - 2) * N * N;
        p.alpha = 2.f; p.cs_beta = -1.f;
        p.M = (int)N; p.K = (int)N; p.L = (int)N;
        p.a_div = 1; p.a_bs1 = S * N * N; p.a_bs2 = 0;
        p.x_div = 1; p.x_bs1 = S * N * N; p.x_bs2 = 0;
        p.o_div = 1; p.o_bs1 = S * N * N; p.o_bs2 = 0;
        p.cs_div = 1; p.cs_bs1 = S * N * N; p.cs_bs2 = 0;
        p.cs_row = N;
        p.kdiv = 1; p.k_lo = N;
        p.qdiv = 0;
        p.o_row = N;
        p.ogdiv = 0;
        p.a_vec = (N % 4 == 0);
        p.x_vec = (N % 4 == 0);
        axis_gemm_launch(p, (int)B, /*is_f32=*/1, stream());
    }
    return out;
}

namespace {
// shared Chebyshev-recurrence GEMM: OUT slot k = 2 * A_slot @ X_slot - C_slot
void cheby_gemm(float* base, long a_off, long x_off, long o_off, long c_off,
                long inst_stride, long B, long N) {
    AxisGemmParams p{};
    p.AT = base + a_off;
    p.X = base + x_off;
    p.OUT = base + o_off;
    p.CSUB = base + c_off;
    p.alpha = 2.f; p.cs_beta = -1.f;
    p.M = (int)N; p.K = (int)N; p.L = (int)N;
    p.a_div = 1; p.a_bs1 = inst_stride; p.a_bs2 = 0;
    p.x_div = 1; p.x_bs1 = inst_stride; p.x_bs2 = 0;
    p.o_div = 1; p.o_bs1 = inst_stride; p.o_bs2 = 0;
    p.cs_div = 1; p.cs_bs1 = inst_stride; p.cs_bs2 = 0;
    p.cs_row = N;
    p.kdiv = 1; p.k_lo = N;
    p.qdiv = 0;
    p.o_row = N;
    p.ogdiv = 0;
    p.a_vec = (N % 4 == 0);
    p.x_vec = (N % 4 == 0);
    axis_gemm_launch(p, (int)B, /*is_f32=*/1, stream());
}
}  // namespace

// Fused dual-random-walk-diffusion support build: fwd series (rowsum-
// normalized P^T polynomials) at slots 0..order, bwd series (colsum-
// normalized, i.e. P_bwd^T in natural orientation) at slots order+1..2order,
// sharing T_0 = I (GCN.py:84-91 semantics).
torch::Tensor dual_rwd_supports(torch::Tensor flow, long order) {
    TORCH_CHECK(flow.is_cuda() && flow.is_contiguous() &&
                flow.scalar_type() == torch::kFloat, "flow must be CUDA f32");
    TORCH_CHECK(flow.dim() == 3 && flow.size(1) == flow.size(2), "flow (B,N,N)");
    TORCH_CHECK(order >= 1, "needs order >= 1");
    const long B = flow.size(0), N = flow.size(1), S = 2 * order + 1;
    TORCH_CHECK(B <= 65535, "too many instances");
    auto out = torch::empty({B, S, N, N}, flow.options());
    auto d = torch::empty({B, N}, flow.options());
    auto dc = torch::empty({B, N}, flow.options());
    const long st = S * N * N;
    float* base = out.data_ptr<float>();
    rwd_rowsum_launch(flow.data_ptr<float>(), d.data_ptr<float>(), B * N,
                      (int)N, stream());
    rwd_colsum_launch(flow.data_ptr<float>(), dc.data_ptr<float>(), B, (int)N,
                      stream());
    // slots 0 (I) and 1 (P_fwd^T)
    rwd_norm_t_launch(flow.data_ptr<float>(), d.data_ptr<float>(), base, B,
                      (int)N, st, stream());
    // slot order+1 (P_bwd^T, natural orientation)
    dual_bwd_norm_launch(flow.data_ptr<float>(), dc.data_ptr<float>(),
                         base + (order + 1) * N * N, B, (int)N, st, stream());
    for (long k = 2; k <= order; ++k)
        cheby_gemm(base, 1 * N * N, (k - 1) * N * N, k * N * N,
                   (k - 2) * N * N, st, B, N);
    for (long k = 2; k <= order; ++k) {
        // bwd_j lives at slot order + j (j >= 1); bwd_0 is the shared I
        const long a_s = order + 1;          // PT_bwd = bwd_1
        const long x_s = order + (k - 1);    // bwd_{k-1}
        const long o_s = order + k;          // bwd_k
        const long c_s = (k == 2) ? 0 : order + (k - 2);
        cheby_gemm(base, a_s * N * N, x_s * N * N, o_s * N * N,
                   c_s * N * N, st, B, N);
    }
    return out;
}

// Fused Chebyshev support build (fixed lambda_max — the reference path,
// since its torch.eig fallback always fires, GCN.py:116-126): rowsum ->
// fused rescaled-Laplacian seed -> recurrence GEMMs.
torch::Tensor cheby_supports(torch::Tensor flow, long order, double lam) {
    TORCH_CHECK(flow.is_cuda() && flow.is_contiguous() &&
                flow.scalar_type() == torch::kFloat, "flow must be CUDA f32");
    TORCH_CHECK(flow.dim() == 3 && flow.size(1) == flow.size(2), "flow (B,N,N)");
    TORCH_CHECK(order >= 1, "needs order >= 1");
    const long B = flow.size(0), N = flow.size(1), S = order + 1;
    TORCH_CHECK(B <= 65535, "too many instances");
    auto out = torch::empty({B, S, N, N}, flow.options());
    auto d = torch::empty({B, N}, flow.options());
    const long st = S * N * N;
    float* base = out.data_ptr<float>();
    rwd_rowsum_launch(flow.data_ptr<float>(), d.data_ptr<float>(), B * N,
                      (int)N, stream());
    cheby_seed_launch(flow.data_ptr<float>(), d.data_ptr<float>(), base, B,
                      (int)N, st, (float)lam, stream());
    for (long k = 2; k <= order; ++k)
        cheby_gemm(base, 1 * N * N, (k - 1) * N * N, k * N * N,
                   (k - 2) * N * N, st, B, N);
    return out;
}

// Fused localpool support build: (B,N,N) -> (B,1,N,N), I + sym_norm(A)
// (GCN.py:71-72) in two launches.
torch::Tensor localpool_supports(torch::Tensor flow) {
    TORCH_CHECK(flow.is_cuda() && flow.is_contiguous() &&
                flow.scalar_type() == torch::kFloat, "flow must be CUDA f32");
    TORCH_CHECK(flow.dim() == 3 && flow.size(1) == flow.size(2), "flow (B,N,N)");
    const long B = flow.size(0), N = flow.size(1);
    auto out = torch::empty({B, 1, N, N}, flow.options());
    auto d = torch::empty({B, N}, flow.options());
    rwd_rowsum_launch(flow.data_ptr<float>(), d.data_ptr<float>(), B * N,
                      (int)N, stream());
    localpool_seed_launch(flow.data_ptr<float>(), d.data_ptr<float>(),
                          out.data_ptr<float>(), B, (int)N, N * N, stream());
    return out;
}

// Fused ReLU backward + bias-grad column sum: dY = dH * 1[Y>0], dbias=colsum.
std::vector<torch::Tensor> relu_bwd_colsum(torch::Tensor dH, torch::Tensor Y,
                                           bool mask) {
    check_in(dH, "dH");
    TORCH_CHECK(dH.scalar_type() == torch::kBFloat16, "bf16 only");
    const long H = dH.size(-1);
    TORCH_CHECK(H >= 8 && (H & (H - 1)) == 0 && H <= 2048,
                "H must be a power of two in [8, 2048]");
    auto dY = torch::empty_like(dH);
    const bool det = at::globalContext().deterministicAlgorithms();
    const long nb = det ? relu_bwd_nblocks(dH.numel()) : 1;
    auto colsum = det ? torch::empty({nb, H}, dH.options().dtype(torch::kFloat))
                      : torch::zeros({H}, dH.options().dtype(torch::kFloat));
    ReluBwdParams p{};
    p.dH = dH.data_ptr();
    p.Y = mask ? Y.data_ptr() : dH.data_ptr();
    p.dY = dY.data_ptr();
    p.colsum = colsum.data_ptr<float>();
    p.total = dH.numel();
    p.H = (int)H;
    p.mask = mask ? 1 : 0;
    p.det = det ? 1 : 0;
    relu_bwd_colsum_launch(p, stream());
    if (det) {
        auto cs = torch::empty({H}, dH.options().dtype(torch::kFloat));
        slab_colsum_launch(colsum.data_ptr<float>(), cs.data_ptr<float>(), nb,
                           H, stream());
        return {dY, cs};
    }
    return {dY, colsum};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("bdgcn_mode1", &bdgcn_mode1, "origin-axis graph product (K1)",
          py::arg("X"), py::arg("GT"), py::arg("id_skip") = false,
          py::arg("no_fill") = false);
    m.def("bdgcn_mode2", &bdgcn_mode2, "dest-axis graph product + bias + act (K2)",
          py::arg("V"), py::arg("A2T"), py::arg("bias"), py::arg("relu"),
          py::arg("N"), py::arg("S"), py::arg("id_skip") = false);
    m.def("bdgcn_mode2_bwd", &bdgcn_mode2_bwd, "backward dV of mode2",
          py::arg("dY"), py::arg("A2"), py::arg("S"), py::arg("id_skip") = false,
          py::arg("no_fill") = false);
    m.def("bdgcn_mode1_bwd", &bdgcn_mode1_bwd, "backward dX of mode1",
          py::arg("dU"), py::arg("A3T"), py::arg("id_skip") = false);
    m.def("row_gemm", &row_gemm, "fused row GEMM + bias + act (K3/K6)");
    m.def("row_gemm_split", &row_gemm_split,
          "row GEMM over split-source rows [XA | XB] (identity-slot-free)");
    m.def("red_gemm", &red_gemm, "fused reduction GEMM X^T@Y + colsum + xdot");
    m.def("red_gemm_split", &red_gemm_split,
          "reduction GEMM over split rows [XA|XB]^T @ [YA|YB]");
    m.def("relu_bwd_colsum", &relu_bwd_colsum, "fused ReLU bwd mask + bias colsum");
    m.def("bdgcn_mode2_fp8", &bdgcn_mode2_fp8, "fp8 e4m3 mode-2 probe (measurement only)");
    m.def("bdgcn_mode1_fp8", &bdgcn_mode1_fp8, "fp8 e4m3 mode-1 probe (measurement only)");
    m.def("bdgcn_mode1_fp8_train", &bdgcn_mode1_fp8_train,
          "fp8-forward mode-1 with bf16 backward twin",
          py::arg("X8"), py::arg("GT8"), py::arg("id_skip") = false,
          py::arg("no_fill") = false);
    m.def("bdgcn_mode2_fp8_train", &bdgcn_mode2_fp8_train,
          "fp8-forward mode-2 + bias + act, bf16 out + fp8 twin");
    m.def("row_gemm_fp8", &row_gemm_fp8, "fp8 projection GEMM");
    m.def("row_gemm_fp8_split", &row_gemm_fp8_split,
          "fp8 projection GEMM over split rows [XA8 | XB8]");
    m.def("rwd_supports", &rwd_supports, "fused random-walk-diffusion support build (K8)");
    m.def("dual_rwd_supports", &dual_rwd_supports, "fused dual-RWD support build");
    m.def("cheby_supports", &cheby_supports, "fused Chebyshev support build");
    m.def("localpool_supports", &localpool_supports, "fused localpool support build");
    m.def("bdgcn_mode2_bwd_fp8", &bdgcn_mode2_bwd_fp8,
          "scaled fp8 gradient contraction dV",
          py::arg("dY8"), py::arg("A28"), py::arg("S"), py::arg("inv_scale"),
          py::arg("dY_bf16"), py::arg("id_skip") = false,
          py::arg("no_fill") = false);
    m.def("bdgcn_mode1_bwd_fp8", &bdgcn_mode1_bwd_fp8,
          "scaled fp8 gradient contraction dX");
    m.def("relu_bwd_colsum_fp8", &relu_bwd_colsum_fp8,
          "ReLU bwd + colsum + fused scaled fp8 dY8");
    m.def("row_gemm_fp8_out_split", &row_gemm_fp8_out_split,
          "split-row row GEMM emitting scaled fp8 only");
    m.def("row_gemm_fp8_out", &row_gemm_fp8_out,
          "row GEMM emitting scaled fp8 only, amax tracked");
    m.def("fp8_scale_update", &fp8_scale_update,
          "delayed-scaling scale/inv update from amax");
    m.def("adam_flat", &adam_flat, "fused flat-buffer Adam step");
    m.def("row_gemm_out", &row_gemm_out, "row GEMM into strided output slice");
    m.def("lstm_step_fwd", &lstm_step_fwd, "fused LSTM cell forward step (K4)");
    m.def("lstm_fused_fwd", &lstm_fused_fwd, "register-resident fused LSTM forward");
    m.def("lstm_fused_bwd", &lstm_fused_bwd, "register-resident fused LSTM backward");
    m.def("lstm_step_bwd", &lstm_step_bwd, "LSTM cell backward pointwise step");
}
