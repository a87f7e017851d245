"""Autograd-integrated ops: HIP kernels on GPU, eager PyTorch on CPU.

The BDGCN layer uses the factored algorithm (mpgcn_amd/ops/eager.py docstring):
    U = mode1(X, Go)            K origin-axis products        [axis_gemm kernel]
    V = U_flat @ Wre            one flat projection GEMM      [row_gemm kernel]
    H = act(mode2(V, Gd) + b)   K dest-axis products, fused   [axis_gemm kernel]

Graph supports carry no gradient (they are built from input data each step,
reference Model_Trainer.py:82-84,106), so backward only produces dX, dW, dbias:
    dY, dbias = relu_bwd_colsum(dH, Y)        [fused mask + bias colsum kernel]
    dV  = mode2_bwd(dY, A2)                   [axis_gemm]
    dW  = (dV^T @ U reordered)                [red_gemm fused reduction kernel]
    dU  = dVflat @ Wre^T                      [row_gemm]
    dX  = mode1_bwd(dU, A3T)                  [axis_gemm]
"""

from __future__ import annotations

import torch

from mpgcn_amd import ops as _ops
from mpgcn_amd.ops import eager

_ROW_GEMM_MAX_N = 128


class GraphOperator:
    """Per-step container for one perspective's support stack in every layout
    the kernels need. Layout permutes are tiny (graph tensors are O(S*N^2))
    and computed lazily, once per training step, shared by all gcn layers and
    by forward+backward.

    Go/Gd: (S, N, N) static or (B, S, N, N) dynamic, in compute dtype.
    """

    def __init__(self, Go: torch.Tensor, Gd: torch.Tensor,
                 id_first: bool | None = None):
        self.Go = Go
        self.Gd = Gd
        self.S = Go.shape[-3]
        self.N = Go.shape[-1]
        if id_first is None:
            id_first = (getattr(Go, "_identity_first", False)
                        and getattr(Gd, "_identity_first", False))
        # identity-support skip: every Chebyshev-family support stack starts
        # with T_0 = I (graph/supports.py tags its outputs), whose products
        # are the inputs themselves — the kernel layouts then EXCLUDE support
        # 0 and the bindings add the identity terms directly (epilogue adds /
        # strided copies), cutting 1/S of every axis contraction's FLOPs and
        # staged bytes. `id_first` is the id_skip flag passed to ext calls.
        self.id_first = bool(id_first) and self.S >= 2
        self._GoT = None
        self._A2T = None
        self._A2 = None
        self._A3T = None
        self._GoT8 = None
        self._A2T8 = None
        self._A28 = None
        self._A3T8 = None

    def _go_k(self) -> torch.Tensor:
        """Origin supports entering the kernels (support 0 dropped in
        id_first mode)."""
        return self.Go.narrow(-3, 1, self.S - 1) if self.id_first else self.Go

    def _gd_k(self) -> torch.Tensor:
        return self.Gd.narrow(-3, 1, self.S - 1) if self.id_first else self.Gd

    @property
    def GoT(self) -> torch.Tensor:
        """GT[..., m, n] = Go[..., n, m] — mode-1 A operand (reduced in
        id_first mode)."""
        if self._GoT is None:
            self._GoT = self._go_k().transpose(-2, -1).contiguous()
        return self._GoT

    @property
    def A2T(self) -> torch.Tensor:
        """A2T[d, c*Se+s] = Gd_k[s, c, d] — mode-2 A operand, (N, N*Se)."""
        if self._A2T is None:
            gd = self._gd_k()
            Se = gd.shape[-3]
            if gd.dim() == 3:
                self._A2T = gd.permute(2, 1, 0).reshape(self.N, self.N * Se).contiguous()
            else:
                B = gd.shape[0]
                self._A2T = gd.permute(0, 3, 2, 1).reshape(B, self.N, self.N * Se).contiguous()
        return self._A2T

    @property
    def GoT8(self) -> torch.Tensor:
        """e4m3 copy of GoT (fp8-forward mode). Supports are normalized
        transition-matrix polynomials, O(1) magnitudes — e4m3 holds them."""
        if self._GoT8 is None:
            self._GoT8 = self.GoT.to(torch.float8_e4m3fn)
        return self._GoT8

    @property
    def A2T8(self) -> torch.Tensor:
        if self._A2T8 is None:
            self._A2T8 = self.A2T.to(torch.float8_e4m3fn)
        return self._A2T8

    @property
    def A28(self) -> torch.Tensor:
        if self._A28 is None:
            self._A28 = self.A2.to(torch.float8_e4m3fn)
        return self._A28

    @property
    def A3T8(self) -> torch.Tensor:
        if self._A3T8 is None:
            self._A3T8 = self.A3T.to(torch.float8_e4m3fn)
        return self._A3T8

    @property
    def A2(self) -> torch.Tensor:
        """A2[c*Se+s, d] = Gd_k[s, c, d] — mode-2 backward A operand."""
        if self._A2 is None:
            gd = self._gd_k()
            Se = gd.shape[-3]
            if gd.dim() == 3:
                self._A2 = gd.permute(1, 0, 2).reshape(self.N * Se, self.N).contiguous()
            else:
                B = gd.shape[0]
                self._A2 = gd.permute(0, 2, 1, 3).reshape(B, self.N * Se, self.N).contiguous()
        return self._A2

    @property
    def A3T(self) -> torch.Tensor:
        """A3T[n, o*N+m] = Go_k[o, n, m] — mode-1 backward A operand."""
        if self._A3T is None:
            go = self._go_k()
            Se = go.shape[-3]
            if go.dim() == 3:
                self._A3T = go.permute(1, 0, 2).reshape(self.N, Se * self.N).contiguous()
            else:
                B = go.shape[0]
                self._A3T = go.permute(0, 2, 1, 3).reshape(B, self.N, Se * self.N).contiguous()
        return self._A3T


_dw_streams: dict = {}


def _dw_stream(device):
    """Side stream for the weight-gradient reduction: red_gemm (dWre) is
    independent of the dU -> dX chain, so it overlaps the rest of the layer
    backward instead of serializing it (the b=32 flagship regime is
    latency-bound on the per-layer chain, not device throughput). Keyed by
    the PRODUCING stream — the branch-overlap forward runs backward on two
    streams, and a single shared side stream would couple their critical
    paths (measured -12%)."""
    key = (device, torch.cuda.current_stream(device))
    s = _dw_streams.get(key)
    if s is None:
        if len(_dw_streams) > 16:
            _dw_streams.clear()
        s = _dw_streams[key] = torch.cuda.Stream(device)
    return s


def _dw_overlapped(ext, dVflat, U2d, dtype, S, C, Hdim):
    """dW via red_gemm on the side stream; returns (dW, join_fn). Caller runs
    the dX chain, then calls join_fn() before returning dW to autograd."""
    if torch.cuda.is_current_stream_capturing():
        # hipGraph capture: run inline (forking a non-capturing stream from
        # inside a capture aborts the process)
        dWreT, _, _ = ext.red_gemm(dVflat, U2d, False, None, 0, 0)
        dWre = dWreT.t().to(dtype)
        dW = dWre.reshape(S, C, S, Hdim).permute(0, 2, 1, 3).reshape(S * S * C, Hdim)
        return dW.contiguous(), (lambda: None)
    cur = torch.cuda.current_stream()
    s = _dw_stream(dVflat.device)
    s.wait_stream(cur)
    with torch.cuda.stream(s):
        dWreT, _, _ = ext.red_gemm(dVflat, U2d, False, None, 0, 0)
        dWre = dWreT.t().to(dtype)
        dW = dWre.reshape(S, C, S, Hdim).permute(0, 2, 1, 3).reshape(S * S * C, Hdim)
        dW = dW.contiguous()
    # under hipGraph capture record_stream is not permitted; the wait edges
    # alone express the dependency there (graph pool owns lifetimes)
    capturing = torch.cuda.is_current_stream_capturing()
    if not capturing:
        dVflat.record_stream(s)
        U2d.record_stream(s)

    def join():
        cur.wait_stream(s)
        if not capturing:
            dW.record_stream(cur)

    return dW, join


def _dw_overlapped_split(ext, dYf, dVred, Xf, Ured, dtype, S, C, Hdim):
    """Split-row variant of _dw_overlapped for the identity-slot-free
    schedule: dWre^T = [dY | dV]^T @ [X | U] via red_gemm_split — the
    identity blocks stream from their source tensors (no slot-0 copies)."""
    def _compute():
        dWreT = ext.red_gemm_split(dYf, dVred, Xf, Ured)
        dWre = dWreT.t().to(dtype)
        dW = dWre.reshape(S, C, S, Hdim).permute(0, 2, 1, 3).reshape(S * S * C, Hdim)
        return dW.contiguous()

    if torch.cuda.is_current_stream_capturing():
        return _compute(), (lambda: None)
    cur = torch.cuda.current_stream()
    s = _dw_stream(dYf.device)
    s.wait_stream(cur)
    with torch.cuda.stream(s):
        dW = _compute()
    for t in (dYf, dVred, Xf, Ured):
        t.record_stream(s)

    def join():
        cur.wait_stream(s)
        dW.record_stream(cur)

    return dW, join


def _row_gemm_chunked(ext, X2d, W, bias, relu):
    """row_gemm with column chunking for N > 128 (e.g. dual-RWD S=5: S*H=160)."""
    R, _ = X2d.shape
    N = W.shape[1]
    if N <= _ROW_GEMM_MAX_N:
        return ext.row_gemm(X2d, W, bias, relu)
    out = torch.empty(R, N, device=X2d.device, dtype=X2d.dtype)
    for n0 in range(0, N, _ROW_GEMM_MAX_N):
        n1 = min(n0 + _ROW_GEMM_MAX_N, N)
        bchunk = bias[n0:n1].contiguous() if bias is not None else None
        ext.row_gemm_out(X2d, W[:, n0:n1].contiguous(), bchunk, relu, out, N, n0)
    return out


class _BDGCNLayerFn(torch.autograd.Function):
    """GPU path of one BDGCN layer via the HIP kernels."""

    @staticmethod
    def forward(ctx, X, W, bias, gop: GraphOperator, relu: bool):
        ext = _ops.get_ext()
        B, N = X.shape[0], X.shape[1]
        C = X.shape[-1]
        S = gop.S
        Hdim = W.shape[1]

        # identity-slot-free schedule: with id_first, mode-1 never needs to
        # MATERIALIZE slot 0 (= X) — the projection and the dW reduction read
        # the identity block straight from X/dY through the split-row kernels,
        # removing the slot_copy read+write entirely (~2.7% of flagship device
        # time). Gated on chunk-aligned part widths and single-call row_gemm N.
        ch = 8 if X.dtype == torch.bfloat16 else 4
        nofill = (gop.id_first and C % ch == 0 and Hdim % ch == 0
                  and S * Hdim <= _ROW_GEMM_MAX_N and S * C <= _ROW_GEMM_MAX_N)
        ctx.nofill = nofill

        U = ext.bdgcn_mode1(X, gop.GoT, gop.id_first, nofill)
        Wre = eager.reorder_projection_weight(W, S, C).contiguous()
        R = B * N * N
        if nofill:  # U: (B,N,N,S-1,C); logical rows are [X | U]
            Vflat = ext.row_gemm_split(X.reshape(R, C),
                                       U.reshape(R, (S - 1) * C),
                                       Wre, None, False)
        else:       # U: (B,N,N,S,C)
            Vflat = _row_gemm_chunked(ext, U.reshape(R, S * C), Wre, None,
                                      False)
        bias_f32 = bias.float().contiguous() if bias is not None else None
        Y = ext.bdgcn_mode2(Vflat.view(B, N, N * S, Hdim), gop.A2T, bias_f32, relu, N, S, gop.id_first)

        ctx.save_for_backward(X, U, Wre, Y)
        ctx.gop = gop
        ctx.relu = relu
        ctx.has_bias = bias is not None
        ctx.dims = (B, N, S, C, Hdim)
        return Y

    @staticmethod
    def backward(ctx, dH):
        ext = _ops.get_ext()
        X, U, Wre, Y = ctx.saved_tensors
        gop: GraphOperator = ctx.gop
        B, N, S, C, Hdim = ctx.dims

        dH = dH.contiguous()
        if dH.dtype == torch.bfloat16 and (Hdim & (Hdim - 1)) == 0 and Hdim >= 8:
            # one fused pass: mask + bias column-sum
            dY, dbias = ext.relu_bwd_colsum(dH, Y, ctx.relu)
            dY = dY.view_as(dH)
            if not ctx.has_bias:
                dbias = None
        else:
            dY = dH * (Y > 0).to(dH.dtype) if ctx.relu else dH
            dbias = dY.sum(dim=(0, 1, 2)).to(torch.float32) if ctx.has_bias else None

        R = B * N * N
        WreT = Wre.t().contiguous()
        if ctx.nofill:
            # reduced dV (no materialized identity-gradient slot); logical
            # rows are [dY | dV] for the dU GEMM and the dW reduction
            dV = ext.bdgcn_mode2_bwd(dY, gop.A2, S, True, True)  # (B,N,N,S-1,H)
            dYf = dY.reshape(R, Hdim)
            dVred = dV.reshape(R, (S - 1) * Hdim)
            dW, join_dw = _dw_overlapped_split(
                ext, dYf, dVred, X.reshape(R, C), U.reshape(R, (S - 1) * C),
                dH.dtype, S, C, Hdim)
            dU = ext.row_gemm_split(dYf, dVred, WreT, None, False)
        else:
            dV = ext.bdgcn_mode2_bwd(dY, gop.A2, S, gop.id_first)  # (B,N,N,S,H)
            dVflat = dV.reshape(R, S * Hdim)
            # dWre^T = dV^T @ U via the fused reduction kernel (f32
            # accumulate), overlapped with the dU -> dX chain on a side stream
            dW, join_dw = _dw_overlapped(ext, dVflat, U.reshape(R, S * C),
                                         dH.dtype, S, C, Hdim)
            dU = _row_gemm_chunked(ext, dVflat, WreT, None, False)
        dX = ext.bdgcn_mode1_bwd(dU.view(B, N, N, S, C), gop.A3T, gop.id_first)
        join_dw()
        db = dbias if ctx.has_bias else None
        return dX, dW, db, None, None


def bdgcn_layer(X, W, bias, gop: GraphOperator, relu: bool = True):
    """One 2-D GCN layer. X: (B,N,N,C); W: (C*S*S, H); bias: (H,) or None."""
    if X.is_cuda:
        return _BDGCNLayerFn.apply(X, W, bias, gop, relu)
    return eager.bdgcn_layer_eager(X, gop.Go, gop.Gd, W, bias, "relu" if relu else "none")


_FP8_MARGIN = 224.0  # half of e4m3 max: headroom over the recorded amax


def make_fp8_state(device) -> dict:
    """Per-layer delayed-scaling state for the fp8 gradient path (dY and dU):
    amax recorded by the producing kernel this step becomes next step's
    quantize scale (device-resident; the whole schedule runs without host
    syncs). amax bootstraps at the margin, i.e. the first step quantizes at
    scale 1 (e4m3's native 2^-9..448 covers typical gradient magnitudes);
    from step 2 the scale tracks the real amax."""
    def triple():
        return {
            "amax": torch.full((1,), _FP8_MARGIN, device=device),
            "scale": torch.ones(1, device=device),
            "inv": torch.ones(1, device=device),
        }

    y, u = triple(), triple()
    return {
        "amax_y": y["amax"], "scale_y": y["scale"], "inv_y": y["inv"],
        "amax_u": u["amax"], "scale_u": u["scale"], "inv_u": u["inv"],
    }


class _BDGCNLayerFp8Fn(torch.autograd.Function):
    """fp8-forward / bf16-backward BDGCN layer.

    The axis contractions are bound by staged BYTES through the per-CU load
    path (profiles/SUMMARY.md: load time ~15x MFMA time per stage), so the
    forward runs every GEMM on e4m3 operands at BK=128 (measured kernel
    levers: 1.43x mode-2, 1.30x mode-1, ~2x on the memory-bound projection).
    fp8 twins flow between ops via dual-write epilogues — U8 feeds the fp8
    projection while U_bf16 is saved for the dW reduction; mode-2 emits the
    bf16 autograd output plus a Y8 twin for the NEXT layer's mode-1, so the
    only standalone quantize in the whole stack is layer 1's LSTM-output cast.
    Backward is byte-for-byte the bf16 path (straight-through estimator
    w.r.t. the weight/activation quantization, standard QAT semantics).
    """

    @staticmethod
    def forward(ctx, X, W, bias, gop: GraphOperator, relu: bool, X8,
                emit_twin: bool, fp8_state: dict):
        import os

        ext = _ops.get_ext()
        B, N = X.shape[0], X.shape[1]
        C = X.shape[-1]
        S = gop.S
        Hdim = W.shape[1]
        if X8 is None:
            X8 = X.to(torch.float8_e4m3fn)
        # identity-slot-free fp8 schedule (see _BDGCNLayerFn): U8/dV carry no
        # materialized identity slot; consumers read X8/dY directly via the
        # split-row kernels. Disabled under MPGCN_FP8_BWD=0 so the bisect
        # path can delegate to _BDGCNLayerFn.backward's fill layout.
        nofill = (gop.id_first and C % 16 == 0 and Hdim % 8 == 0
                  and S * Hdim <= _ROW_GEMM_MAX_N and S * C <= _ROW_GEMM_MAX_N
                  and os.environ.get("MPGCN_FP8_BWD", "1") != "0")
        # U stays fp8-ONLY: mode-1 writes half the bf16 path's output bytes
        # (U is the step's largest tensor), and backward's dW reduction reads
        # the fp8 U directly (red_gemm y_fp8 staging)
        U8 = ext.bdgcn_mode1_fp8_train(X8, gop.GoT8, gop.id_first, nofill)
        Wre = eager.reorder_projection_weight(W, S, C).contiguous()
        Wre8 = Wre.to(torch.float8_e4m3fn)
        R = B * N * N
        if nofill:  # U8: (B,N,N,S-1,C); logical rows are [X8 | U8]
            V8 = ext.row_gemm_fp8_split(X8.reshape(R, C),
                                        U8.reshape(R, (S - 1) * C), Wre8)
        else:
            V8 = ext.row_gemm_fp8(U8.reshape(R, S * C), Wre8)
        bias_f32 = bias.float().contiguous() if bias is not None else None
        Y, Y8 = ext.bdgcn_mode2_fp8_train(
            V8.view(B, N, N * S, Hdim), gop.A2T8, bias_f32, relu, N, S,
            emit_twin, gop.id_first,
        )
        # X saved first (and ctx.nofill pinned False) so the MPGCN_FP8_BWD=0
        # bisect path can delegate to _BDGCNLayerFn.backward, which unpacks
        # (X, U, Wre, Y); nofill mode additionally saves X8 for the dW split
        # (bisect is env-gated off there, so the layouts never mix)
        if nofill:
            ctx.save_for_backward(X, X8, U8, Wre, Y)
        else:
            ctx.save_for_backward(X, U8, Wre, Y)
        ctx.nofill = False
        ctx.nofill_fp8 = nofill
        ctx.gop = gop
        ctx.relu = relu
        ctx.has_bias = bias is not None
        ctx.dims = (B, N, S, C, Hdim)
        ctx.fp8_state = fp8_state
        if emit_twin:
            ctx.mark_non_differentiable(Y8)
            return Y, Y8
        return Y

    @staticmethod
    def backward(ctx, dH, _dY8=None):
        import os

        if os.environ.get("MPGCN_FP8_BWD", "1") == "0" and not ctx.nofill_fp8:
            # debug/bisect: fp8 forward with the full bf16 backward (red_gemm
            # reads the saved U8 directly either way). Forward disables the
            # identity-slot-free layout under this env, so the ctx layouts
            # match; a mid-run env flip keeps the normal fp8 backward instead.
            dX, dW, db, _, _ = _BDGCNLayerFn.backward(ctx, dH)
            return dX, dW, db, None, None, None, None, None
        # Scaled-fp8 gradient contractions: gradients live well below e4m3's
        # 2^-9 subnormal floor, so every quantize carries a DEVICE-resident
        # dynamic scale and the axis-kernel epilogue descales — no host sync.
        # Quantization is FUSED into the producers with one-step-DELAYED
        # per-layer scales (a standalone amax+mul+cast chain, and even a
        # same-step amax pre-pass, cost more memory traffic than the fp8
        # contraction saves — both measured):
        #   dY8: written by relu_bwd_colsum in its streaming pass, which also
        #        records this step's amax for the next step's scale;
        #   dU8: the ONLY dU materialization — row_gemm emits scaled fp8
        #        directly (its sole consumer is the fp8 dX contraction).
        # Weight-gradient reductions (red_gemm) stay bf16-accumulated-f32.
        ext = _ops.get_ext()
        if ctx.nofill_fp8:
            _X, X8s, U8, Wre, Y = ctx.saved_tensors
        else:
            _X, U8, Wre, Y = ctx.saved_tensors
        gop: GraphOperator = ctx.gop
        st = ctx.fp8_state
        B, N, S, C, Hdim = ctx.dims

        dH = dH.contiguous()
        ext.fp8_scale_update(st["amax_y"], st["scale_y"], st["inv_y"], _FP8_MARGIN)
        dY, dY8, dbias = ext.relu_bwd_colsum_fp8(dH, Y, ctx.relu,
                                                 st["scale_y"], st["amax_y"])
        if not ctx.has_bias:
            dbias = None

        R = B * N * N
        WreT = Wre.t().contiguous()
        ext.fp8_scale_update(st["amax_u"], st["scale_u"], st["inv_u"], _FP8_MARGIN)
        if ctx.nofill_fp8:
            # identity-slot-free: reduced dV; logical rows [dY | dV] / the
            # dW reduction reads [X8 | U8] — numerics identical to the fill
            # layout (the fills wrote exactly dY and X8)
            dV = ext.bdgcn_mode2_bwd_fp8(dY8, gop.A28, S, st["inv_y"], dY,
                                         True, True)  # (B,N,N,S-1,H)
            dYf = dY.reshape(R, Hdim)
            dVred = dV.reshape(R, (S - 1) * Hdim)
            dW, join_dw = _dw_overlapped_split(
                ext, dYf, dVred, X8s.reshape(R, C),
                U8.reshape(R, (S - 1) * C), dH.dtype, S, C, Hdim)
            dU8 = ext.row_gemm_fp8_out_split(dYf, dVred, WreT,
                                             st["scale_u"], st["amax_u"])
        else:
            dV = ext.bdgcn_mode2_bwd_fp8(dY8, gop.A28, S, st["inv_y"], dY,
                                         gop.id_first)
            dVflat = dV.reshape(R, S * Hdim)
            dW, join_dw = _dw_overlapped(ext, dVflat, U8.reshape(R, S * C),
                                         dH.dtype, S, C, Hdim)
            dU8 = ext.row_gemm_fp8_out(dVflat, WreT,
                                       st["scale_u"], st["amax_u"])
        dX = ext.bdgcn_mode1_bwd_fp8(dU8.view(B, N, N, S, C), gop.A3T8,
                                     st["inv_u"], gop.id_first)
        join_dw()
        return dX, dW, dbias, None, None, None, None, None


def fp8_forward_compatible(N: int, C: int, Hdim: int, S: int) -> bool:
    """Shape gate for the vector-only fp8 tiles (ext.hip contracts)."""
    return ((N * C) % 256 == 0 and C % 16 == 0 and (N * Hdim) % 256 == 0
            and Hdim % 16 == 0 and (S * C) % 16 == 0 and S * Hdim <= 128)


def bdgcn_layer_fp8(X, W, bias, gop: GraphOperator, relu: bool = True,
                    X8=None, emit_twin: bool = True, fp8_state: dict = None):
    """fp8-forward BDGCN layer: (Y_bf16, Y8_twin), or Y_bf16 alone with
    emit_twin=False (last layer — its consumer is the bf16 FC head).
    fp8_state: per-layer make_fp8_state() dict (delayed dU scaling); a
    transient one is created when omitted (single-shot calls/tests).
    GPU-only; callers gate on fp8_forward_compatible and fall back to
    bdgcn_layer otherwise."""
    if not X.is_cuda:
        raise RuntimeError("fp8-forward mode requires a GPU")
    if fp8_state is None:
        fp8_state = make_fp8_state(X.device)
    return _BDGCNLayerFp8Fn.apply(X, W, bias, gop, relu, X8, emit_twin,
                                  fp8_state)


class _Mode1ProjFn(torch.autograd.Function):
    """mode-1 + projection half of a BDGCN layer (region-partition path,
    mpgcn_amd/parallel/region.py): the layer splits at the all-to-all seam, so
    each half is its own autograd node. X may be destination-sharded
    (B, N, Nd, C) with Nd = N/P; the axis kernels take rectangular shapes."""

    @staticmethod
    def forward(ctx, X, W, gop: GraphOperator):
        ext = _ops.get_ext()
        B, No, Nd, C = X.shape
        S = gop.S
        Hdim = W.shape[1]
        U = ext.bdgcn_mode1(X, gop.GoT, gop.id_first)  # (B, No, Nd, S, C)
        Wre = eager.reorder_projection_weight(W, S, C).contiguous()
        Vflat = _row_gemm_chunked(ext, U.reshape(B * No * Nd, S * C), Wre, None, False)
        ctx.save_for_backward(U, Wre)
        ctx.gop = gop
        ctx.dims = (B, No, Nd, S, C, Hdim)
        return Vflat.view(B, No, Nd, S * Hdim)

    @staticmethod
    def backward(ctx, dV):
        ext = _ops.get_ext()
        U, Wre = ctx.saved_tensors
        B, No, Nd, S, C, Hdim = ctx.dims
        R = B * No * Nd
        dVflat = dV.reshape(R, S * Hdim).contiguous()
        dWreT, _, _ = ext.red_gemm(dVflat, U.reshape(R, S * C), False, None, 0, 0)
        dWre = dWreT.t().to(dV.dtype)
        dW = dWre.reshape(S, C, S, Hdim).permute(0, 2, 1, 3).reshape(S * S * C, Hdim)
        dU = _row_gemm_chunked(ext, dVflat, Wre.t().contiguous(), None, False)
        dX = ext.bdgcn_mode1_bwd(dU.view(B, No, Nd, S, C), ctx.gop.A3T, ctx.gop.id_first)
        return dX, dW, None


class _Mode2BiasActFn(torch.autograd.Function):
    """mode-2 + bias + activation half of a BDGCN layer (region-partition
    path). V is origin-sharded (B, Nm, N, S, H) with Nm = N/P."""

    @staticmethod
    def forward(ctx, V, bias, gop: GraphOperator, relu: bool):
        ext = _ops.get_ext()
        B, Nm, N, S, Hdim = V.shape
        bias_f32 = bias.float().contiguous() if bias is not None else None
        Y = ext.bdgcn_mode2(V.reshape(B, Nm, N * S, Hdim).contiguous(),
                            gop.A2T, bias_f32, relu, N, S, gop.id_first)
        ctx.save_for_backward(Y)
        ctx.gop = gop
        ctx.relu = relu
        ctx.has_bias = bias is not None
        ctx.dims = (B, Nm, N, S, Hdim)
        return Y  # (B, Nm, N, H)

    @staticmethod
    def backward(ctx, dH):
        ext = _ops.get_ext()
        (Y,) = ctx.saved_tensors
        B, Nm, N, S, Hdim = ctx.dims
        dH = dH.contiguous()
        if dH.dtype == torch.bfloat16 and (Hdim & (Hdim - 1)) == 0 and Hdim >= 8:
            dY, dbias = ext.relu_bwd_colsum(dH, Y, ctx.relu)
            dY = dY.view_as(dH)
            if not ctx.has_bias:
                dbias = None
        else:
            dY = dH * (Y > 0).to(dH.dtype) if ctx.relu else dH
            dbias = dY.sum(dim=(0, 1, 2)).to(torch.float32) if ctx.has_bias else None
        dV = ext.bdgcn_mode2_bwd(dY, ctx.gop.A2, S, ctx.gop.id_first)  # (B, Nm, N, S, H)
        return dV, dbias if ctx.has_bias else None, None, None


def mode1_proj(X, W, gop: GraphOperator):
    """mode-1 contraction + projection: (B, N, Nd, C) -> (B, N, Nd, S*H)."""
    if X.is_cuda:
        return _Mode1ProjFn.apply(X, W, gop)
    S = gop.S
    C = X.shape[-1]
    B, No, Nd = X.shape[:3]
    U = eager.mode1_apply(X, gop.Go)
    Wre = eager.reorder_projection_weight(W, S, C)
    return (U.reshape(B * No * Nd, S * C) @ Wre).view(B, No, Nd, -1)


def mode2_bias_act(V, bias, gop: GraphOperator, relu: bool = True):
    """mode-2 contraction + bias + act: (B, Nm, N, S, H) -> (B, Nm, N, H)."""
    if V.is_cuda:
        return _Mode2BiasActFn.apply(V, bias, gop, relu)
    if gop.Gd.dim() == 3:
        Y = torch.einsum("scd,bmcsh->bmdh", gop.Gd.to(V.dtype), V)
    else:
        Y = torch.einsum("bscd,bmcsh->bmdh", gop.Gd.to(V.dtype), V)
    if bias is not None:
        Y = Y + bias.to(Y.dtype)
    return torch.relu(Y) if relu else Y


class _FusedLSTMLastFn(torch.autograd.Function):
    """Single-layer LSTM over R sequences, returning only the LAST hidden state
    (the only timestep MPGCN consumes, reference MPGCN.py:104). Input dim 1.

    x: (R, T) compute-dtype; weights in torch nn.LSTM layout:
    w_ih (4H, 1), w_hh (4H, H), b_ih (4H,), b_hh (4H,).
    """

    @staticmethod
    def forward(ctx, x, w_ih, w_hh, b_ih, b_hh):
        ext = _ops.get_ext()
        R, T = x.shape
        Hd = w_hh.shape[1]
        dev = x.device
        whh = w_hh.contiguous()
        wih_f = w_ih.reshape(-1).float().contiguous()
        bias_f = (b_ih.float() + b_hh.float()).contiguous()

        # T-slab state buffers: H_buf[t] is h BEFORE step t (h_0 = 0), so the
        # whole-sequence weight-grad reduction in backward is ONE red_gemm pass
        # over (T*R) rows instead of T separate kernel sweeps.
        H_buf = torch.zeros(T + 1, R, Hd, device=dev, dtype=x.dtype)
        C_buf = torch.zeros(T + 1, R, Hd, device=dev, dtype=torch.float32)
        G_buf = torch.empty(T, R, 4 * Hd, device=dev, dtype=x.dtype)
        xc = x.contiguous()
        for t in range(T):
            ext.lstm_step_fwd(xc, T, t, H_buf[t], C_buf[t], whh, wih_f, bias_f,
                              H_buf[t + 1], C_buf[t + 1], G_buf[t])
        ctx.save_for_backward(xc, whh, wih_f, H_buf, C_buf, G_buf)
        ctx.T = T
        return H_buf[T]

    @staticmethod
    def backward(ctx, dh_last):
        ext = _ops.get_ext()
        xc, whh, wih_f, H_buf, C_buf, G_buf = ctx.saved_tensors
        T = ctx.T
        R = xc.shape[0]
        Hd = whh.shape[1]
        dev = xc.device

        dh = dh_last.contiguous()
        dc = None
        dc_buf = [torch.empty(R, Hd, device=dev, dtype=torch.float32) for _ in range(2)]
        dG_buf = torch.empty(T, R, 4 * Hd, device=dev, dtype=xc.dtype)
        for t in range(T - 1, -1, -1):
            dc_out = dc_buf[t % 2]
            ext.lstm_step_bwd(dh, dc, G_buf[t], C_buf[t], C_buf[t + 1],
                              dG_buf[t], dc_out)
            dc = dc_out
            if t > 0:
                dh = ext.row_gemm(dG_buf[t], whh, None, False)  # dgates @ W_hh

        # one fused reduction over all T steps: dW_hh = dG^T @ h_prev,
        # dbias = colsum(dG), dw_ih = dG^T x
        xT = xc.t().contiguous()  # (T, R): row t*R+r aligns with dG/h slabs
        dwhh, dbias, dwih = ext.red_gemm(
            dG_buf.view(T * R, 4 * Hd), H_buf[:T].reshape(T * R, Hd),
            True, xT.view(-1, 1), 1, 0,
        )
        need_dx = ctx.needs_input_grad[0]
        dx = None
        if need_dx:
            wih_c = wih_f.to(xc.dtype).view(-1, 1).contiguous()
            dx_all = ext.row_gemm(dG_buf.view(T * R, 4 * Hd), wih_c, None, False)
            dx = dx_all.view(T, R).t().contiguous()
        wdt = whh.dtype
        return (
            dx,
            dwih.view(-1, 1).to(wdt),
            dwhh.to(wdt),
            dbias,
            dbias.clone(),
        )


class _RegLSTMFn(torch.autograd.Function):
    """Register-resident LSTM (bf16, H=32), ANY sequence length via chunking:
    T is processed in chunks of <= 8 steps whose h/c states live entirely in
    registers; chunk boundaries checkpoint (h, c) — O(R*H) per boundary —
    and the backward walks the chunks in reverse, recomputing each chunk's
    forward in-kernel from its entry checkpoint and chaining (dh, dc)
    through the boundaries. T <= 8 (the reference's use case) is exactly the
    round-1 single-chunk schedule with zero checkpoint traffic; T = 14 no
    longer falls off the slab-path cliff (round-1: 2.3x slower per sample)."""

    @staticmethod
    def forward(ctx, x, w_ih, w_hh, b_ih, b_hh, T=None):
        ext = _ops.get_ext()
        whh = w_hh.contiguous()
        wih_f = w_ih.reshape(-1).float().contiguous()
        bias_f = (b_ih.float() + b_hh.float()).contiguous()
        T = x.shape[1] if T is None else T
        n_chunks = (T + 7) // 8
        width = n_chunks * 8
        ctx.pre_padded = x.shape[1] == width and width != T
        if ctx.pre_padded:
            # caller already laid x out at the kernel's padded row width
            # (models/mpgcn.py fuses cast+permute+pad into one copy) — no
            # extra pad pass here
            xp = x.contiguous()
        else:
            pad = width - T
            xp = (torch.nn.functional.pad(x, (0, pad)).contiguous()
                  if pad else x.contiguous())
        h = c = None
        starts = []  # (h, c) entering each chunk; chunk 0 enters at zeros
        for ci in range(n_chunks):
            Tc = min(8, T - ci * 8)
            last = ci == n_chunks - 1
            starts.append((h, c))
            h, c = ext.lstm_fused_fwd(xp, ci * 8, Tc, whh, wih_f, bias_f,
                                      h, c, not last)
        ctx.save_for_backward(
            xp, whh, wih_f, bias_f,
            *[t for hc in starts[1:] for t in hc],  # chunk-entry checkpoints
        )
        ctx.T = T
        ctx.n_chunks = n_chunks
        return h

    @staticmethod
    def backward(ctx, dh):
        ext = _ops.get_ext()
        xp, whh, wih_f, bias_f, *ckpts = ctx.saved_tensors
        T, n_chunks = ctx.T, ctx.n_chunks
        need_dx = ctx.needs_input_grad[0]
        whhT = whh.t().contiguous()
        dx = torch.empty_like(xp) if need_dx else None
        dh = dh.contiguous()
        dc = None
        dwhh = dbias = dwih = None
        for ci in range(n_chunks - 1, -1, -1):
            Tc = min(8, T - ci * 8)
            if ci > 0:
                h0, c0 = ckpts[2 * (ci - 1)], ckpts[2 * (ci - 1) + 1]
            else:
                h0 = c0 = None
            dwhh_i, dbias_i, dwih_i, dh, dc = ext.lstm_fused_bwd(
                xp, ci * 8, Tc, whh, whhT, wih_f, bias_f, dh,
                h0, c0, dc, ci > 0, dx,
            )
            dwhh = dwhh_i if dwhh is None else dwhh + dwhh_i
            dbias = dbias_i if dbias is None else dbias + dbias_i
            dwih = dwih_i if dwih is None else dwih + dwih_i
        wdt = whh.dtype
        if need_dx and ctx.pre_padded:
            dx[:, T:].zero_()  # grad must match the padded input layout
        return (
            (dx if ctx.pre_padded else dx[:, :T].contiguous())
            if need_dx else None,
            dwih.view(-1, 1).to(wdt),
            dwhh.to(wdt),
            dbias,
            dbias.clone(),
            None,
        )


def fused_lstm_last(x, w_ih, w_hh, b_ih, b_hh, T=None):
    """Last hidden state of a 1-layer batch-first LSTM over (R, T) scalar
    inputs. T (optional) gives the logical sequence length when x rows are
    pre-padded to the chunk width (multiple of 8) — the flagship path fuses
    cast+permute+pad into one copy in models/mpgcn.py and passes T here."""
    T_log = x.shape[1] if T is None else T
    Hd = w_hh.shape[1]
    if x.is_cuda and Hd == 32 and x.dtype == torch.bfloat16:
        # register-resident chunked schedule covers any T (chunks of <= 8)
        return _RegLSTMFn.apply(x, w_ih, w_hh, b_ih, b_hh, T_log)
    xl = x if T_log == x.shape[1] else x[:, :T_log]
    kernel_ok = Hd == 32 or (Hd == 16 and x.dtype == torch.float32)
    if x.is_cuda and kernel_ok:
        return _FusedLSTMLastFn.apply(xl.contiguous(), w_ih, w_hh, b_ih, b_hh)
    out, _, _ = eager.lstm_forward_eager(xl.unsqueeze(-1), w_ih, w_hh, b_ih, b_hh)
    return out[:, -1, :]


class _LinearActFn(torch.autograd.Function):
    """Fused Linear(+bias)+ReLU head via row_gemm; backward in plain torch
    (the FC head is (R,32)->(R,1), a trivial fraction of step time)."""

    @staticmethod
    def forward(ctx, X2d, weight, bias, relu):
        ext = _ops.get_ext()
        Wt = weight.t().contiguous()  # (in, out)
        bias_f = bias.float().contiguous() if bias is not None else None
        out = _row_gemm_chunked(ext, X2d, Wt, bias_f, relu)
        ctx.save_for_backward(X2d, weight, out)
        ctx.relu = relu
        ctx.has_bias = bias is not None
        return out

    @staticmethod
    def backward(ctx, dOut):
        X2d, weight, out = ctx.saved_tensors
        dY = (dOut * (out > 0).to(dOut.dtype) if ctx.relu else dOut).contiguous()
        if weight.shape[0] == 1:
            # rank-1 head (the MPGCN FC): the outer product is a broadcast
            # multiply — rocBLAS spends ~250us + a workspace memset on it
            dX = dY * weight.to(dY.dtype).reshape(1, -1)
        else:
            dX = dY @ weight.to(dY.dtype)
        # dW = dY^T @ X via the fused reduction kernel (rocBLAS is ~25x off
        # roofline on this 1-column tall reduction); colsum(dY) gives dbias
        ext = _ops.get_ext()
        dWt, db_cs, _ = ext.red_gemm(dY, X2d, ctx.has_bias, None, 0, 0)
        dW = dWt.view(weight.shape).to(weight.dtype)
        db = db_cs if ctx.has_bias else None
        return dX, dW, db, None


def linear_act(X2d, weight, bias, relu: bool = True):
    """act(X @ weight^T + bias) with torch nn.Linear weight layout (out, in)."""
    if X2d.is_cuda:
        return _LinearActFn.apply(X2d, weight, bias, relu)
    out = torch.nn.functional.linear(X2d, weight, bias)
    return torch.relu(out) if relu else out
