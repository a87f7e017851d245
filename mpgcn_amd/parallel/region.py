"""Region partition: shard the N x N OD activation grid across GPUs.

The reference is single-device; at 4096 regions a single
(B, N, N, C) activation is ~4 GB and the per-layer intermediates several times
that, so the activation grid itself must shard (weights are tiny and stay
replicated — the model is activation-heavy).

Scheme (the all-to-all formulation of SURVEY.md §7 "hard parts"):
  * activations live DESTINATION-sharded: X_p = X[:, :, d_p : d_p + N/P, :]
    — each rank holds all origin rows for a slice of destination columns;
  * mode-1 (contract the ORIGIN axis) is then fully local;
  * the projection GEMM is row-local;
  * before mode-2 (contract the DESTINATION axis) an all-to-all re-shards
    from destination-sharded to ORIGIN-sharded (each rank: all destinations
    for a slice of origin rows) — mode-2 becomes local;
  * a second all-to-all re-shards back for the next layer's mode-1.
  Per BDGCN layer: 2 all-to-alls of O(B*N^2*C/P) bytes per rank over xGMI —
  the bandwidth-critical collective of this workload.

The LSTM/FC stages are pointwise over (origin, destination) pairs and run on
any sharding. Losses are computed shard-locally; weight gradients all-reduce
through the usual GradAllReducer (weights replicated). The all-to-alls are
autograd-aware (backward = the inverse all-to-all), so the whole sharded
forward trains with plain autograd.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from mpgcn_amd.ops import (
    GraphOperator,
    fused_lstm_last,
    linear_act,
    mode1_proj,
    mode2_bias_act,
)


class _AllToAllShard(torch.autograd.Function):
    """Differentiable all-to-all that converts a destination-sharded tensor
    (B, N, Nl, F) into an origin-sharded one (B, Nl, N, F) or back.

    Forward 'd2o': input  (B, N, N/P, F)  ->  output (B, N/P, N, F)
    Forward 'o2d': input  (B, N/P, N, F)  ->  output (B, N, N/P, F)
    Backward is the opposite direction (all-to-all is self-adjoint up to the
    permutation).
    """

    @staticmethod
    def forward(ctx, x, direction: str, group):
        ctx.direction = direction
        ctx.group = group
        return _a2a(x, direction, group)

    @staticmethod
    def backward(ctx, g):
        inv = "o2d" if ctx.direction == "d2o" else "d2o"
        return _a2a(g.contiguous(), inv, ctx.group), None, None


def _a2a_exchange(send: torch.Tensor, group) -> torch.Tensor:
    """all_to_all_single with a CPU-staged fallback for gloo+CUDA (gloo's
    CUDA transport covers all_reduce/broadcast but not all-to-all) — lets
    the region path run multi-rank on shared/heterogeneous setups; RCCL
    takes the direct device path."""
    if send.is_cuda and dist.get_backend(group) == "gloo":
        send_c = send.cpu()
        recv_c = torch.empty_like(send_c)
        dist.all_to_all_single(recv_c, send_c, group=group)
        return recv_c.to(send.device, non_blocking=True)
    recv = torch.empty_like(send)
    dist.all_to_all_single(recv, send, group=group)
    return recv


def _a2a(x: torch.Tensor, direction: str, group) -> torch.Tensor:
    P = dist.get_world_size(group)
    if direction == "d2o":
        B, N, Nl, F = x.shape
        assert N == Nl * P, (N, Nl, P)
        # send chunk q = origin rows [q*Nl, (q+1)*Nl) of the local dest slice
        send = x.reshape(B, P, Nl, Nl, F).permute(1, 0, 2, 3, 4).contiguous()
        recv = _a2a_exchange(send, group)
        # recv[p] = origin rows (local) x dest cols of peer p
        out = recv.permute(1, 2, 0, 3, 4).reshape(B, Nl, N, F)
        return out.contiguous()
    else:
        B, Nl, N, F = x.shape
        assert N == Nl * P, (N, Nl, P)
        send = x.reshape(B, Nl, P, Nl, F).permute(2, 0, 1, 3, 4).contiguous()
        recv = _a2a_exchange(send, group)
        out = recv.permute(1, 0, 2, 3, 4).reshape(B, N, Nl, F)
        return out.contiguous()


def dest_to_origin(x, group=None):
    """(B, N, N/P, F) destination-sharded -> (B, N/P, N, F) origin-sharded."""
    return _AllToAllShard.apply(x, "d2o", group)


def origin_to_dest(x, group=None):
    """(B, N/P, N, F) origin-sharded -> (B, N, N/P, F) destination-sharded."""
    return _AllToAllShard.apply(x, "o2d", group)


def shard_dest(x_full: torch.Tensor, rank: int, P: int) -> torch.Tensor:
    """Slice the destination axis of a full (B, ..., N, N, F) tensor."""
    N = x_full.shape[-2]
    Nl = N // P
    return x_full[..., rank * Nl:(rank + 1) * Nl, :].contiguous()


class _ShardedFp8Layer(torch.autograd.Function):
    """One sharded BDGCN layer in fp8-forward mode, fused across the
    exchange seams: mode1+proj (destination-sharded, fp8 kernels) ->
    all-to-all of the fp8 V8 -> mode2+bias+act (origin-sharded) ->
    all-to-all of the fp8 Y8 twin -> dequant. BOTH forward exchanges carry
    1-BYTE payloads — the all-to-all is this workload's bandwidth-critical
    collective, so fp8 halves it. The transported Y8 bytes
    double as the next layer's X8 operand (zero extra quantization), exactly
    like the square path's twin chaining. Backward transports bf16 gradients
    through the inverse all-to-alls and reuses the square path's scaled-fp8
    gradient machinery (delayed per-layer scales in `st`).
    The LAST layer instead transports bf16 Y (its consumer is the bf16 FC
    head — parity with the square path's emit_twin=False)."""

    @staticmethod
    def forward(ctx, X, W, bias, gop: GraphOperator, relu: bool, X8,
                group, st: dict, last: bool):
        from mpgcn_amd.ops import eager
        from mpgcn_amd.ops.functional import _ops

        ext = _ops.get_ext()
        B, N, Nl, C = X.shape
        S = gop.S
        Hd = W.shape[1]
        if X8 is None:
            X8 = X.to(torch.float8_e4m3fn)
        U8 = ext.bdgcn_mode1_fp8_train(X8, gop.GoT8, gop.id_first)
        Wre = eager.reorder_projection_weight(W, S, C).contiguous()
        Wre8 = Wre.to(torch.float8_e4m3fn)
        V8 = ext.row_gemm_fp8(U8.reshape(B * N * Nl, S * C), Wre8)
        Vo8 = _a2a(V8.view(B, N, Nl, S * Hd).view(torch.uint8), "d2o",
                   group).view(torch.float8_e4m3fn)
        bias_f32 = bias.float().contiguous() if bias is not None else None
        Y, Y8 = ext.bdgcn_mode2_fp8_train(
            Vo8.view(B, Nl, N * S, Hd).contiguous(), gop.A2T8, bias_f32,
            relu, N, S, not last, gop.id_first,
        )  # Y: (B, Nl, N, Hd) origin-sharded
        if last:
            Xd = _a2a(Y, "o2d", group)
            Xd8 = Y8  # undefined tensor (not emitted)
        else:
            Xd8 = _a2a(Y8.view(torch.uint8), "o2d",
                       group).view(torch.float8_e4m3fn)
            Xd = Xd8.to(torch.bfloat16)
        ctx.save_for_backward(U8, Wre, Y)
        ctx.gop = gop
        ctx.relu = relu
        ctx.has_bias = bias is not None
        ctx.dims = (B, N, Nl, S, C, Hd)
        ctx.group = group
        ctx.st = st
        if last:
            return Xd
        ctx.mark_non_differentiable(Xd8)
        return Xd, Xd8

    @staticmethod
    def backward(ctx, dXd, _dXd8=None):
        from mpgcn_amd.ops.functional import _FP8_MARGIN, _ops

        ext = _ops.get_ext()
        U8, Wre, Y = ctx.saved_tensors
        gop: GraphOperator = ctx.gop
        st = ctx.st
        B, N, Nl, S, C, Hd = ctx.dims
        # inverse of the Y exchange: dest-sharded grad -> origin-sharded
        dY = _a2a(dXd.contiguous(), "d2o", ctx.group).contiguous()
        ext.fp8_scale_update(st["amax_y"], st["scale_y"], st["inv_y"], _FP8_MARGIN)
        dYm, dY8, dbias = ext.relu_bwd_colsum_fp8(dY, Y, ctx.relu,
                                                  st["scale_y"], st["amax_y"])
        if not ctx.has_bias:
            dbias = None
        dV = ext.bdgcn_mode2_bwd_fp8(dY8.view(B, Nl, N, Hd), gop.A28, S,
                                     st["inv_y"], dYm.view(B, Nl, N, Hd),
                                     gop.id_first)  # (B, Nl, N, S, Hd)
        # inverse of the V exchange: origin-sharded grad -> dest-sharded
        dVd = _a2a(dV.reshape(B, Nl, N, S * Hd), "o2d", ctx.group)
        R = B * N * Nl
        dVflat = dVd.reshape(R, S * Hd).contiguous()
        dWreT, _, _ = ext.red_gemm(dVflat, U8.reshape(R, S * C), False, None, 0, 0)
        dWre = dWreT.t().to(dXd.dtype)
        dW = dWre.reshape(S, C, S, Hd).permute(0, 2, 1, 3).reshape(S * S * C, Hd)
        ext.fp8_scale_update(st["amax_u"], st["scale_u"], st["inv_u"], _FP8_MARGIN)
        dU8 = ext.row_gemm_fp8_out(dVflat, Wre.t().contiguous(),
                                   st["scale_u"], st["amax_u"])
        dX = ext.bdgcn_mode1_bwd_fp8(dU8.view(B, N, Nl, S, C), gop.A3T8,
                                     st["inv_u"], gop.id_first)
        return dX, dW, dbias, None, None, None, None, None, None


def region_fp8_compatible(N: int, P: int, C: int, Hd: int, S: int) -> bool:
    """Shape gate for the fp8 sharded layer (vector-only fp8 tiles on the
    shard extents Nl = N/P)."""
    Nl = N // P
    return ((Nl * C) % 256 == 0 and C % 16 == 0 and (Nl * Hd) % 256 == 0
            and Hd % 16 == 0 and (S * C) % 16 == 0 and S * Hd <= 128)


def bdgcn_layer_sharded(Xd, gop: GraphOperator, W, bias, group=None, relu=True):
    """One BDGCN layer on a destination-sharded input.

    Xd: (B, N, N/P, C) destination-sharded; gop: GraphOperator over the full
    (replicated) supports — graphs are O(S*N^2); W: (C*S*S, H).
    Returns the next layer's destination-sharded input (B, N, N/P, H).

    On GPU the two local halves run through the HIP axis kernels (the
    bindings accept rectangular origin/destination extents); on CPU the same
    math runs as einsums. Math identical to
    mpgcn_amd.ops.eager.bdgcn_layer_eager (verified by tests/test_region.py
    against the unsharded computation).
    """
    S = gop.S
    B, N, Nl, C = Xd.shape
    Hdim = W.shape[1]

    # mode-1 (contract the full, local origin axis) + row-local projection
    V = mode1_proj(Xd, W, gop)  # (B, N, Nl, S*H)
    # re-shard: destination-sharded -> origin-sharded (full dest axis)
    Vo = dest_to_origin(V, group)  # (B, Nl, N, S*H)
    # mode-2: contract the (full, local) destination axis, + bias + act
    Y = mode2_bias_act(Vo.view(B, Nl, N, S, Hdim), bias, gop, relu)
    # re-shard back for the next layer's mode-1
    return origin_to_dest(Y, group)  # (B, N, Nl, H)


def mpgcn_forward_sharded(model, x_seq_shard, G_list, group=None):
    """Full MPGCN forward on destination-sharded inputs.

    model: an mpgcn_amd.models.MPGCN (weights replicated across ranks);
    x_seq_shard: (B, T, N, N/P, 1) — this rank's destination slice;
    G_list: full graphs, the usual [static (S,N,N), (O_dyn, D_dyn)] contract.
    Returns the destination-sharded prediction (B, 1, N, N/P, 1).
    """
    B, T, N, Nl, _ = x_seq_shard.shape
    P = dist.get_world_size(group)
    if N % P != 0:
        raise ValueError(
            f"region partition needs the region count ({N}) divisible by the "
            f"world size ({P}); pad the grid or change the rank count")
    fp8 = bool(getattr(model, "fp8_forward", False)) and x_seq_shard.is_cuda
    if fp8 and not region_fp8_compatible(
            N, P, model.lstm_hidden_dim,
            model.branch_models[0]["spatial"][0].hidden_dim, model.K):
        import warnings

        warnings.warn(
            "fp8_forward shard shapes incompatible (region_fp8_compatible); "
            "running the region shards in bf16", stacklevel=2)
        fp8 = False
    gops = model._graph_operators(G_list)
    cd = model.compute_dtype
    if x_seq_shard.is_cuda:
        # one fused cast+transpose+pad copy at the LSTM kernel's row width
        # (same as models/mpgcn.py forward)
        Tpad = -(-T // 8) * 8
        lstm_in = torch.empty(B * N * Nl, Tpad, dtype=cd,
                              device=x_seq_shard.device)
        lv = lstm_in.view(B, N, Nl, Tpad)
        lv[..., :T].copy_(x_seq_shard.squeeze(-1).permute(0, 2, 3, 1))
        if Tpad != T:
            lv[..., T:].zero_()
    else:
        lstm_in = (
            x_seq_shard.to(cd).permute(0, 2, 3, 1, 4)
            .reshape(B * N * Nl, T).contiguous()
        )
    outs = []
    for m in range(model.M):
        branch = model.branch_models[m]
        h = fused_lstm_last(
            lstm_in,
            branch["temporal"].weight_ih_l0.to(cd),
            branch["temporal"].weight_hh_l0.to(cd),
            branch["temporal"].bias_ih_l0,
            branch["temporal"].bias_hh_l0,
            T=T,
        )
        X = h.reshape(B, N, Nl, model.lstm_hidden_dim)
        gop = gops[m]
        if fp8:
            from mpgcn_amd.ops.functional import make_fp8_state

            X8 = None
            n_sp = len(branch["spatial"])
            for i, layer in enumerate(branch["spatial"]):
                st = getattr(layer, "_fp8_state", None)
                if st is None or st["amax_u"].device != X.device:
                    st = layer._fp8_state = make_fp8_state(X.device)
                last = i + 1 == n_sp
                out = _ShardedFp8Layer.apply(
                    X, layer.W.to(cd), layer.b, gop, layer.relu, X8,
                    group, st, last,
                )
                if last:
                    X = out
                else:
                    X, X8 = out
        else:
            for layer in branch["spatial"]:
                X = bdgcn_layer_sharded(
                    X, gop, layer.W.to(cd), layer.b, group, relu=layer.relu
                )
        fc = branch["fc"][0]
        out = linear_act(X.reshape(B * N * Nl, -1), fc.weight.to(cd), fc.bias, True)
        outs.append(out.view(B, N, Nl, 1))
    stacked = torch.stack(outs, dim=-1)
    if model.fusion == "attention":
        # learned softmax fusion is pointwise over (origin, destination) pairs,
        # so it shards trivially; fusion_w is replicated and its gradient
        # all-reduces through the usual GradAllReducer like every other weight
        w = torch.softmax(model.fusion_w, dim=0).to(stacked.dtype)
        ens = (stacked * w).sum(dim=-1)
    else:
        ens = torch.mean(stacked, dim=-1)
    return ens.float().unsqueeze(1)
