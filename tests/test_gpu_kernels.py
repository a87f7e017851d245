"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference of
the same op (run on an MI355X via gpurun; skipped without a GPU)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _ext():
    from mpgcn_amd import ops

    assert ops.has_ext(), "HIP extension must be present on a GPU box"
    return ops.get_ext()


def _mode1_ref(X, G):
    if G.dim() == 3:
        return torch.einsum("onm,bndl->bmdol", G, X)
    return torch.einsum("bonm,bndl->bmdol", G, X)


def _mode2_ref(V, G):
    if G.dim() == 3:
        return torch.einsum("scd,bmcsh->bmdh", G, V)
    return torch.einsum("bscd,bmcsh->bmdh", G, V)


def _tol(dtype):
    return dict(atol=1e-3, rtol=1e-3) if dtype == torch.float32 else dict(atol=8e-2, rtol=8e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("N,dyn", [(64, False), (47, False), (64, True), (33, True)])
def test_mode1(dtype, N, dyn):
    ext = _ext()
    torch.manual_seed(0)
    B, S, C = 2, 3, 32
    X = torch.randn(B, N, N, C, device=DEV).to(dtype)
    G = torch.randn(B, S, N, N, device=DEV).to(dtype) if dyn else torch.randn(S, N, N, device=DEV).to(dtype)
    U = ext.bdgcn_mode1(X.contiguous(), G.transpose(-2, -1).contiguous())
    ref = _mode1_ref(X.float(), G.float())
    assert U.shape == (B, N, N, S, C)
    torch.testing.assert_close(U.float(), ref, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("N,S,dyn,relu", [(64, 3, False, True), (47, 3, True, False), (32, 5, False, True)])
def test_mode2(dtype, N, S, dyn, relu):
    ext = _ext()
    torch.manual_seed(1)
    B, H = 2, 32
    V = torch.randn(B, N, N, S, H, device=DEV).to(dtype)
    G = torch.randn(B, S, N, N, device=DEV).to(dtype) if dyn else torch.randn(S, N, N, device=DEV).to(dtype)
    bias = torch.randn(H, device=DEV)
    if dyn:
        A2T = G.permute(0, 3, 2, 1).reshape(B, N, N * S).contiguous()
    else:
        A2T = G.permute(2, 1, 0).reshape(N, N * S).contiguous()
    Y = ext.bdgcn_mode2(V.reshape(B, N, N * S, H).contiguous(), A2T, bias, relu, N, S)
    ref = _mode2_ref(V.float(), G.float()) + bias
    if relu:
        ref = torch.relu(ref)
    torch.testing.assert_close(Y.float(), ref, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_mode2_bwd(dtype):
    ext = _ext()
    torch.manual_seed(2)
    B, N, S, H = 2, 48, 3, 32
    dY = torch.randn(B, N, N, H, device=DEV).to(dtype)
    G = torch.randn(S, N, N, device=DEV).to(dtype)
    A2 = G.permute(1, 0, 2).reshape(N * S, N).contiguous()
    dV = ext.bdgcn_mode2_bwd(dY.contiguous(), A2, S)
    # dV[b,m,c,s,h] = sum_d G[s,c,d] dY[b,m,d,h]
    ref = torch.einsum("scd,bmdh->bmcsh", G.float(), dY.float())
    torch.testing.assert_close(dV.float(), ref, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("dyn", [False, True])
def test_mode1_bwd(dtype, dyn):
    ext = _ext()
    torch.manual_seed(3)
    B, N, S, C = 2, 40, 3, 32
    dU = torch.randn(B, N, N, S, C, device=DEV).to(dtype)
    G = torch.randn(B, S, N, N, device=DEV).to(dtype) if dyn else torch.randn(S, N, N, device=DEV).to(dtype)
    if dyn:
        A3T = G.permute(0, 2, 1, 3).reshape(B, N, S * N).contiguous()
        ref = torch.einsum("bonm,bmdol->bndl", G.float(), dU.float())
    else:
        A3T = G.permute(1, 0, 2).reshape(N, S * N).contiguous()
        ref = torch.einsum("onm,bmdol->bndl", G.float(), dU.float())
    dX = ext.bdgcn_mode1_bwd(dU.contiguous(), A3T)
    torch.testing.assert_close(dX.float(), ref, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("R,K,Nc,relu", [(1000, 96, 96, False), (513, 32, 1, True), (256, 128, 32, False)])
def test_row_gemm(dtype, R, K, Nc, relu):
    ext = _ext()
    torch.manual_seed(4)
    X = torch.randn(R, K, device=DEV).to(dtype)
    W = torch.randn(K, Nc, device=DEV).to(dtype)
    bias = torch.randn(Nc, device=DEV)
    out = ext.row_gemm(X.contiguous(), W.contiguous(), bias, relu)
    ref = X.float() @ W.float() + bias
    if relu:
        ref = torch.relu(ref)
    torch.testing.assert_close(out.float(), ref, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("R,K,Nc", [(5000, 96, 96), (2048, 128, 32), (777, 160, 160)])
def test_red_gemm(dtype, R, K, Nc):
    ext = _ext()
    torch.manual_seed(11)
    X = (torch.randn(R, K, device=DEV) * 0.1).to(dtype)
    Y = (torch.randn(R, Nc, device=DEV) * 0.1).to(dtype)
    xv = torch.randn(R, 3, device=DEV).to(dtype)
    out, colsum, xdot = ext.red_gemm(X.contiguous(), Y.contiguous(), True,
                                     xv.contiguous(), 3, 1)
    ref = X.float().t() @ Y.float()
    rel = (out - ref).norm() / ref.norm()
    assert rel < (1e-5 if dtype == torch.float32 else 1e-2), rel
    torch.testing.assert_close(colsum, X.float().sum(0), atol=1e-1, rtol=1e-2)
    ref_xdot = (X.float() * xv[:, 1].float().unsqueeze(1)).sum(0)
    torch.testing.assert_close(xdot, ref_xdot, atol=1e-1, rtol=2e-2)


@pytest.mark.parametrize("mask", [True, False])
def test_relu_bwd_colsum(mask):
    ext = _ext()
    torch.manual_seed(13)
    B, N, H = 3, 40, 32
    dH = torch.randn(B, N, N, H, device=DEV, dtype=torch.bfloat16)
    Y = torch.randn(B, N, N, H, device=DEV, dtype=torch.bfloat16)
    dY, colsum = ext.relu_bwd_colsum(dH, Y, mask)
    ref = dH.float() * (Y.float() > 0) if mask else dH.float()
    torch.testing.assert_close(dY.float(), ref.view_as(dY.float()), atol=1e-3, rtol=1e-3)
    torch.testing.assert_close(colsum, ref.reshape(-1, H).sum(0), atol=0.5, rtol=1e-2)


def test_row_gemm_chunked_wide():
    from mpgcn_amd.ops.functional import _row_gemm_chunked

    ext = _ext()
    torch.manual_seed(5)
    R, K, Nc = 500, 160, 160  # S=5 dual-RWD shape: S*C = S*H = 160 > 128
    X = torch.randn(R, K, device=DEV, dtype=torch.bfloat16)
    W = torch.randn(K, Nc, device=DEV, dtype=torch.bfloat16)
    out = _row_gemm_chunked(ext, X.contiguous(), W.contiguous(), None, False)
    ref = X.float() @ W.float()
    torch.testing.assert_close(out.float(), ref, atol=8e-2, rtol=8e-2)


@pytest.mark.parametrize("dtype,T", [(torch.float32, 7), (torch.bfloat16, 7),
                                      (torch.float32, 10),  # slab path
                                      (torch.bfloat16, 10),  # 2-chunk path
                                      (torch.bfloat16, 16)])  # 2 full chunks
def test_lstm_forward_vs_nn_lstm(dtype, T):
    from mpgcn_amd.ops.functional import fused_lstm_last

    torch.manual_seed(6)
    R, H = 333, 32
    x = torch.randn(R, T, device=DEV)
    lstm = torch.nn.LSTM(1, H, num_layers=1, batch_first=True).to(DEV)
    ref, _ = lstm(x.unsqueeze(-1), (torch.zeros(1, R, H, device=DEV),
                                    torch.zeros(1, R, H, device=DEV)))
    out = fused_lstm_last(
        x.to(dtype), lstm.weight_ih_l0.to(dtype), lstm.weight_hh_l0.to(dtype),
        lstm.bias_ih_l0, lstm.bias_hh_l0,
    )
    torch.testing.assert_close(out.float(), ref[:, -1, :].float(), **_tol(dtype))


def test_lstm_backward_grads_match_autograd():
    from mpgcn_amd.ops import eager
    from mpgcn_amd.ops.functional import fused_lstm_last

    torch.manual_seed(7)
    R, T, H = 200, 6, 32
    x = torch.randn(R, T, device=DEV)
    w_ih = torch.randn(4 * H, 1, device=DEV) * 0.2
    w_hh = torch.randn(4 * H, H, device=DEV) * 0.2
    b_ih = torch.randn(4 * H, device=DEV) * 0.1
    b_hh = torch.randn(4 * H, device=DEV) * 0.1

    # autograd reference through the eager fp32 implementation
    ref_in = [t.clone().requires_grad_(True) for t in (x, w_ih, w_hh, b_ih, b_hh)]
    out, _, _ = eager.lstm_forward_eager(ref_in[0].unsqueeze(-1), *ref_in[1:])
    loss = out[:, -1, :].square().sum()
    loss.backward()

    ins = [t.clone().requires_grad_(True) for t in (x, w_ih, w_hh, b_ih, b_hh)]
    h = fused_lstm_last(*ins)
    h.square().sum().backward()

    for got, ref, name in zip(ins, ref_in, ("x", "w_ih", "w_hh", "b_ih", "b_hh")):
        torch.testing.assert_close(
            got.grad.float(), ref.grad.float(), atol=2e-2, rtol=2e-2, msg=name
        )


@pytest.mark.parametrize("T", [7, 14, 16])
def test_reg_lstm_bf16_grads_match_autograd(T):
    """The register-resident bf16 LSTM (forward + recompute-backward) vs the
    eager fp32 autograd reference. T > 8 exercises the CHUNKED schedule
    (boundary checkpoints + chained (dh, dc) across chunk backwards)."""
    from mpgcn_amd.ops import eager
    from mpgcn_amd.ops.functional import _RegLSTMFn

    torch.manual_seed(12)
    R, H = 3000, 32
    x = torch.randn(R, T, device=DEV) * 0.5
    w_ih = torch.randn(4 * H, 1, device=DEV) * 0.2
    w_hh = torch.randn(4 * H, H, device=DEV) * 0.2
    b_ih = torch.randn(4 * H, device=DEV) * 0.1
    b_hh = torch.randn(4 * H, device=DEV) * 0.1

    ref_in = [t.clone().requires_grad_(True) for t in (x, w_ih, w_hh, b_ih, b_hh)]
    out, _, _ = eager.lstm_forward_eager(ref_in[0].unsqueeze(-1), *ref_in[1:])
    out[:, -1, :].square().sum().backward()

    ins = [x.bfloat16(), w_ih.bfloat16(), w_hh.bfloat16(), b_ih, b_hh]
    ins = [t.clone().requires_grad_(True) for t in ins]
    h = _RegLSTMFn.apply(*ins)
    torch.testing.assert_close(h.float(), out[:, -1, :].detach(), atol=5e-2, rtol=5e-2)
    h.float().square().sum().backward()

    for got, ref, name in zip(ins, ref_in, ("x", "w_ih", "w_hh", "b_ih", "b_hh")):
        err = (got.grad.float() - ref.grad).norm() / (ref.grad.norm() + 1e-8)
        assert err < 3e-2, f"{name}: rel grad err {err:.4f}"


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_gcn1d_hip_path_matches_eager(dtype):
    """1-D GCN (K5) on the HIP kernels (mode-1 axis GEMM + fused row_gemm)
    vs the eager einsum reference."""
    from mpgcn_amd.models.gcn1d import GCN

    torch.manual_seed(3)
    K, C, Hd, Bn, N_ = 3, 8, 16, 4, 48
    m = GCN(K=K, input_dim=C, hidden_dim=Hd).to(DEV)
    G = torch.rand(K, N_, N_, device=DEV)
    x = torch.randn(Bn, N_, C, device=DEV)
    ref = m.cpu()(G.cpu(), x.cpu())  # eager fallback path
    m = m.to(DEV)
    got = m(G.to(dtype), x.to(dtype))
    torch.testing.assert_close(got.float().cpu(), ref, **_tol(dtype))


def test_fused_localpool_supports_match_torch_path():
    from mpgcn_amd.graph import build_supports

    torch.manual_seed(2)
    flow = torch.rand(5, 47, 47) * 4
    flow[0, 3] = 0
    ref = build_supports(flow, "localpool", 1)
    got = build_supports(flow.to(DEV), "localpool", 1)
    torch.testing.assert_close(got.cpu(), ref, atol=1e-5, rtol=1e-5)


@pytest.mark.parametrize("kernel", ["random_walk_diffusion",
                                    "dual_random_walk_diffusion", "chebyshev"])
@pytest.mark.parametrize("order", [1, 2, 4])
@pytest.mark.parametrize("N_", [47, 256])
def test_fused_supports_match_torch_path(kernel, order, N_):
    """The fused K8 support builds (rowsum/colsum + fused seed kernels +
    alpha/CSUB Chebyshev GEMMs) vs the stock torch chain on CPU, for all
    three fused kernel types."""
    from mpgcn_amd.graph import build_supports

    torch.manual_seed(order * 31 + N_)
    flow = torch.rand(5, N_, N_) * 4
    flow[0, 3] = 0  # empty row: rcp/rsqrt guards
    flow[1, :, 5] = 0  # empty column: dual-RWD backward-series guard
    ref = build_supports(flow, kernel, order)  # CPU torch
    got = build_supports(flow.to(DEV), kernel, order)  # fused
    assert getattr(got, "_identity_first", False)
    torch.testing.assert_close(got.cpu(), ref, atol=2e-4, rtol=2e-4)


@pytest.mark.parametrize("dyn", [False, True])
def test_identity_skip_matches_full_path(dyn):
    """id_first mode (support 0 = I skipped: reduced operand layouts,
    epilogue identity adds, strided identity copies) must reproduce the full
    contraction bit-for... closely: the only difference is the f32 epilogue
    add of the exact identity term vs an MFMA against a quantized-to-bf16 I
    (which IS exact for I), so forward and all grads match tightly."""
    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.ops import GraphOperator, bdgcn_layer

    torch.manual_seed(9)
    S, C, Hd, Bn, N_ = 3, 32, 32, 2, 64
    X = torch.randn(Bn, N_, N_, C, device=DEV).bfloat16()
    nb = Bn if dyn else 1
    Go = build_supports(torch.rand(nb, N_, N_, device=DEV),
                        "random_walk_diffusion", S - 1)
    Gd = build_supports(torch.rand(nb, N_, N_, device=DEV),
                        "random_walk_diffusion", S - 1)
    if not dyn:
        Go, Gd = Go.squeeze(0), Gd.squeeze(0)
    Go, Gd = Go.bfloat16().contiguous(), Gd.bfloat16().contiguous()
    W = (0.1 * torch.randn(C * S * S, Hd, device=DEV)).bfloat16()
    b = 0.05 * torch.randn(Hd, device=DEV)

    def run(id_first):
        Xi = X.clone().requires_grad_(True)
        Wi = W.clone().requires_grad_(True)
        bi = b.clone().requires_grad_(True)
        gop = GraphOperator(Go, Gd, id_first=id_first)
        Y = bdgcn_layer(Xi, Wi, bi, gop, relu=True)
        Y.square().sum().backward()
        return Y, Xi.grad, Wi.grad, bi.grad

    Yf, dXf, dWf, dbf = run(False)
    Ys, dXs, dWs, dbs = run(True)
    torch.testing.assert_close(Ys, Yf, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(dXs.float(), dXf.float(), atol=1e-1, rtol=5e-2)
    torch.testing.assert_close(dWs.float(), dWf.float(), atol=2.0, rtol=5e-2)
    torch.testing.assert_close(dbs.float(), dbf.float(), atol=2.0, rtol=5e-2)


def test_identity_skip_fp8_matches_full():
    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.ops import GraphOperator, bdgcn_layer_fp8

    torch.manual_seed(10)
    S, C, Hd, Bn, N_ = 3, 32, 32, 2, 256
    X = torch.randn(Bn, N_, N_, C, device=DEV).bfloat16()
    Go = build_supports(torch.rand(1, N_, N_, device=DEV),
                        "random_walk_diffusion", S - 1).squeeze(0)
    Gd = build_supports(torch.rand(1, N_, N_, device=DEV),
                        "random_walk_diffusion", S - 1).squeeze(0)
    Go, Gd = Go.bfloat16().contiguous(), Gd.bfloat16().contiguous()
    W = (0.1 * torch.randn(C * S * S, Hd, device=DEV)).bfloat16()

    def run(id_first):
        gop = GraphOperator(Go, Gd, id_first=id_first)
        Y, _ = bdgcn_layer_fp8(X, W, None, gop, relu=True)
        return Y

    Yf = run(False)
    Ys = run(True)
    rel = (Ys.float() - Yf.float()).norm() / (Yf.float().norm() + 1e-9)
    assert rel < 0.03, rel.item()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("dyn", [False, True])
def test_bdgcn_layer_fwd_bwd_vs_eager(dtype, dyn):
    from mpgcn_amd.ops import GraphOperator, bdgcn_layer, eager

    torch.manual_seed(8)
    B, N, C, H, S = 2, 40, 32, 32, 3
    X = torch.randn(B, N, N, C, device=DEV).to(dtype)
    if dyn:
        Go = torch.randn(B, S, N, N, device=DEV).to(dtype) * 0.3
        Gd = torch.randn(B, S, N, N, device=DEV).to(dtype) * 0.3
    else:
        Go = torch.randn(S, N, N, device=DEV).to(dtype) * 0.3
        Gd = torch.randn(S, N, N, device=DEV).to(dtype) * 0.3
    W = (torch.randn(C * S * S, H, device=DEV) * 0.05).to(dtype)
    b = torch.randn(H, device=DEV)

    Xg = X.clone().requires_grad_(True)
    Wg = W.clone().requires_grad_(True)
    bg = b.clone().requires_grad_(True)
    out = bdgcn_layer(Xg, Wg, bg, GraphOperator(Go, Gd), relu=True)
    out.square().sum().backward()

    Xr = X.float().clone().requires_grad_(True)
    Wr = W.float().clone().requires_grad_(True)
    br = b.clone().requires_grad_(True)
    ref = eager.bdgcn_layer_eager(Xr, Go.float(), Gd.float(), Wr, br, "relu")
    ref.square().sum().backward()

    torch.testing.assert_close(out.float(), ref.detach(), **_tol(dtype))
    if dtype == torch.float32:
        tol = dict(atol=5e-2, rtol=5e-2)
        torch.testing.assert_close(Xg.grad.float(), Xr.grad, **tol)
        torch.testing.assert_close(Wg.grad.float(), Wr.grad, **tol)
        torch.testing.assert_close(bg.grad.float(), br.grad, **tol)
    else:
        # bf16 grads compound rounding through dY->dV->dW; elementwise bounds
        # are the wrong criterion — check normalized Frobenius error instead
        for got, ref_g in ((Xg.grad, Xr.grad), (Wg.grad, Wr.grad), (bg.grad, br.grad)):
            err = (got.float() - ref_g).norm() / ref_g.norm()
            assert err < 2e-2, f"relative grad error {err:.4f}"


def test_full_model_gpu_vs_cpu_oracle():
    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.models import MPGCN

    torch.manual_seed(9)
    B, N, K, H, T = 2, 32, 3, 32, 7
    model_cpu = MPGCN(M=2, K=K, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                      gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N)
    model_gpu = MPGCN(M=2, K=K, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                      gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N,
                      compute_dtype=torch.bfloat16).to(DEV)
    model_gpu.load_state_dict(model_cpu.state_dict())

    x = torch.rand(B, T, N, N, 1)
    flow = torch.rand(B, N, N)
    Gs = build_supports(torch.rand(1, N, N), "random_walk_diffusion", K - 1)[0]
    Go = build_supports(flow, "random_walk_diffusion", K - 1)
    Gd = build_supports(flow.transpose(-2, -1), "random_walk_diffusion", K - 1)

    with torch.no_grad():
        ref = model_cpu(x, [Gs, (Go, Gd)])
        out = model_gpu(x.to(DEV), [Gs.to(DEV), (Go.to(DEV), Gd.to(DEV))]).cpu()
    torch.testing.assert_close(out, ref, atol=5e-2, rtol=5e-2)


def test_train_step_decreases_loss_on_gpu():
    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.models import MPGCN

    torch.manual_seed(10)
    B, N, K, H, T = 4, 32, 3, 32, 7
    model = MPGCN(M=2, K=K, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                  gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N,
                  compute_dtype=torch.bfloat16).to(DEV)
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    x = torch.rand(B, T, N, N, 1, device=DEV)
    y = torch.rand(B, 1, N, N, 1, device=DEV) + 1.0
    flow = torch.rand(B, N, N, device=DEV)
    Gs = build_supports(torch.rand(1, N, N, device=DEV), "random_walk_diffusion", K - 1)[0]
    Go = build_supports(flow, "random_walk_diffusion", K - 1)
    Gd = build_supports(flow.transpose(-2, -1), "random_walk_diffusion", K - 1)

    losses = []
    for _ in range(30):
        out = model(x, [Gs, (Go, Gd)])
        loss = torch.nn.functional.mse_loss(out, y)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.7, losses[:3] + losses[-3:]


def test_full_model_dual_rwd_s5_gpu():
    """S=5 (dual random-walk) exercises the chunked row_gemm path (S*H=160)
    inside a real train step on GPU."""
    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.models import MPGCN

    torch.manual_seed(14)
    B, N, H, T, order = 2, 32, 32, 7, 2
    S = 2 * order + 1
    model = MPGCN(M=2, K=S, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                  gcn_hidden_dim=H, gcn_num_layers=2, num_nodes=N,
                  compute_dtype=torch.bfloat16).to(DEV)
    model_cpu = MPGCN(M=2, K=S, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                      gcn_hidden_dim=H, gcn_num_layers=2, num_nodes=N)
    model_cpu.load_state_dict(model.state_dict())

    x = torch.rand(B, T, N, N, 1)
    flow = torch.rand(B, N, N) + 0.05
    Gs = build_supports(torch.rand(1, N, N) + 0.05, "dual_random_walk_diffusion", order)[0]
    Go = build_supports(flow, "dual_random_walk_diffusion", order)
    Gd = build_supports(flow.transpose(-2, -1), "dual_random_walk_diffusion", order)

    ref = model_cpu(x, [Gs, (Go, Gd)])
    out = model(x.to(DEV), [Gs.to(DEV), (Go.to(DEV), Gd.to(DEV))])
    torch.testing.assert_close(out.cpu(), ref, atol=6e-2, rtol=6e-2)

    loss = out.square().sum()
    loss.backward()
    for n, p in model.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_deterministic_mode_bitwise_reproducible():
    """torch.use_deterministic_algorithms(True) routes the atomic reductions
    through per-block workspaces: identical seeds give bit-identical training
    (SURVEY §5's reproducibility-as-race-detector obligation)."""
    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.models import MPGCN

    torch.use_deterministic_algorithms(True)
    try:
        results = []
        for _ in range(2):
            torch.manual_seed(21)
            torch.cuda.manual_seed_all(21)
            model = MPGCN(M=2, K=3, input_dim=1, lstm_hidden_dim=32,
                          lstm_num_layers=1, gcn_hidden_dim=32, gcn_num_layers=3,
                          num_nodes=48, compute_dtype=torch.bfloat16).to(DEV)
            opt = torch.optim.Adam(model.parameters(), lr=1e-3)
            x = torch.rand(4, 7, 48, 48, 1, device=DEV)
            y = torch.rand(4, 1, 48, 48, 1, device=DEV)
            flow = torch.rand(4, 48, 48, device=DEV)
            Gs = build_supports(torch.rand(1, 48, 48, device=DEV),
                                "random_walk_diffusion", 2)[0]
            Go = build_supports(flow, "random_walk_diffusion", 2)
            Gd = build_supports(flow.transpose(-2, -1), "random_walk_diffusion", 2)
            losses = []
            for _ in range(8):
                loss = torch.nn.functional.mse_loss(model(x, [Gs, (Go, Gd)]), y)
                opt.zero_grad()
                loss.backward()
                opt.step()
                losses.append(loss.item())
            results.append(losses)
        assert results[0] == results[1], (results[0][-1], results[1][-1])
    finally:
        torch.use_deterministic_algorithms(False)


@pytest.mark.parametrize("id_first", [False, True])
def test_rectangular_sharded_halves_match_eager(id_first):
    """Region-partition path on GPU: the layer split at the all-to-all seam
    (mode1_proj | mode2_bias_act) on rectangular shards — the HIP bindings'
    rectangular origin/destination extents — matches the fp32 eager layer,
    forward and backward, with the re-shard emulated by slice/cat.
    id_first=True exercises the identity-support skip on RECTANGULAR
    extents (the path the region trainer takes with build_supports graphs)."""
    from mpgcn_amd.ops import GraphOperator, eager, mode1_proj, mode2_bias_act

    torch.manual_seed(11)
    Nn, S, C, Hd, B, P = 96, 3, 32, 32, 2, 4
    Nl = Nn // P
    dt = torch.bfloat16
    X32 = torch.randn(B, Nn, Nn, C, device=DEV)
    Go = torch.randn(S, Nn, Nn, device=DEV) / Nn**0.5
    Gd = torch.randn(S, Nn, Nn, device=DEV) / Nn**0.5
    if id_first:  # the skip's contract: support 0 IS the identity
        eye = torch.eye(Nn, device=DEV)
        Go[0] = eye
        Gd[0] = eye
    W32 = (torch.randn(C * S * S, Hd, device=DEV) / (C * S * S) ** 0.5).requires_grad_()
    bias32 = torch.randn(Hd, device=DEV, requires_grad=True)

    ref_in = X32.clone().requires_grad_()
    ref = eager.bdgcn_layer_eager(ref_in, Go, Gd, W32, bias32, "relu")
    ref.square().sum().backward()

    X = X32.to(dt).requires_grad_()
    W = W32.detach().to(dt).requires_grad_()
    bias = bias32.detach().clone().requires_grad_()
    gop = GraphOperator(Go.to(dt), Gd.to(dt), id_first=id_first)
    Vs = [mode1_proj(X[:, :, p * Nl:(p + 1) * Nl, :].contiguous(), W, gop)
          for p in range(P)]
    Vfull = torch.cat(Vs, dim=2)  # emulated all-to-all
    Ys = []
    for p in range(P):
        Vo = Vfull[:, p * Nl:(p + 1) * Nl].reshape(B, Nl, Nn, S, Hd)
        Ys.append(mode2_bias_act(Vo, bias, gop, True))
    out = torch.cat(Ys, dim=1)

    def relerr(a, b):
        return (a.float() - b).norm() / b.norm().clamp_min(1e-6)

    assert relerr(out, ref) < 3e-2, relerr(out, ref)
    out.float().square().sum().backward()
    assert relerr(X.grad, ref_in.grad) < 4e-2
    assert relerr(W.grad, W32.grad) < 4e-2
    assert relerr(bias.grad, bias32.grad) < 4e-2
    torch.cuda.synchronize()


@pytest.mark.timeout(600)
def test_capacity_4096_regions_train_step():
    """Config #5 scale (BASELINE.json: 4096-region OD) on one GPU: a full
    train step at N=4096, batch 1 — exercises the >1e9-element axis-kernel
    index ranges (offsets are 64-bit throughout) and the 288 GB HBM sizing.
    Numerics at this scale are pinned by a layer comparison at N=2048."""
    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.models import MPGCN
    from mpgcn_amd.ops import GraphOperator, bdgcn_layer, eager

    # layer numerics at N=2048 (same code path, 8x less eager work)
    torch.manual_seed(12)
    Nn, S, C, Hd = 2048, 3, 32, 32
    X32 = torch.randn(1, Nn, Nn, C, device=DEV)
    Go = torch.randn(S, Nn, Nn, device=DEV) / Nn**0.5
    Gd = torch.randn(S, Nn, Nn, device=DEV) / Nn**0.5
    W32 = torch.randn(C * S * S, Hd, device=DEV) / (C * S * S) ** 0.5
    b32 = torch.randn(Hd, device=DEV)
    ref = eager.bdgcn_layer_eager(X32, Go, Gd, W32, b32, "relu")
    gop = GraphOperator(Go.bfloat16(), Gd.bfloat16())
    out = bdgcn_layer(X32.bfloat16(), W32.bfloat16(), b32, gop, True)
    rel = (out.float() - ref).norm() / ref.norm()
    assert rel < 3e-2, rel.item()
    del X32, Go, Gd, ref, out, gop
    torch.cuda.empty_cache()

    # full model step at N=4096
    N4 = 4096
    model = MPGCN(M=2, K=3, input_dim=1, lstm_hidden_dim=32, lstm_num_layers=1,
                  gcn_hidden_dim=32, gcn_num_layers=3, num_nodes=N4,
                  compute_dtype=torch.bfloat16).to(DEV)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    x = torch.rand(1, 7, N4, N4, 1, device=DEV)
    y = torch.rand(1, 1, N4, N4, 1, device=DEV)
    flow = torch.rand(1, N4, N4, device=DEV)
    Gs = build_supports(torch.rand(1, N4, N4, device=DEV),
                        "random_walk_diffusion", 2)[0]
    Go4 = build_supports(flow, "random_walk_diffusion", 2)
    Gd4 = build_supports(flow.transpose(-2, -1), "random_walk_diffusion", 2)
    losses = []
    for _ in range(3):
        loss = torch.nn.functional.mse_loss(model(x, [Gs, (Go4, Gd4)]), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert losses[-1] < losses[0], losses
    peak = torch.cuda.max_memory_allocated() / 2**30
    assert peak < 200, f"peak {peak:.1f} GiB"


def test_fp8_mode2_probe_numerics():
    """fp8 e4m3 probe path (measurement-only, docs/ROADMAP.md): same axis
    engine at BK=128 on OCP fp8 operands; numerics within e4m3 quantization
    error of the fp32 reference."""
    ext = _ext()
    torch.manual_seed(5)
    B, Nn, S, Hd = 4, 256, 3, 32
    V32 = torch.randn(B, Nn, Nn * S, Hd, device=DEV) / (Nn * S) ** 0.25
    A32 = torch.randn(Nn, Nn * S, device=DEV) / (Nn * S) ** 0.25
    ref = torch.einsum("dk,bmkh->bmdh", A32, V32)
    y8 = ext.bdgcn_mode2_fp8(V32.to(torch.float8_e4m3fn),
                             A32.to(torch.float8_e4m3fn), None, False, Nn, S)
    rel = (y8.float() - ref).norm() / ref.norm()
    assert rel < 8e-2, rel.item()


@pytest.mark.timeout(300)
def test_rectangular_sharded_halves_1024_p8():
    """Region partition at the config #5 aspect ratio (8-way shards of a
    large grid): N=1024, P=8 — Nl=128-wide rectangular extents through the
    HIP kernels, forward vs the fp32 eager layer."""
    from mpgcn_amd.ops import GraphOperator, eager, mode1_proj, mode2_bias_act

    torch.manual_seed(13)
    Nn, S, C, Hd, B, P = 1024, 3, 32, 32, 1, 8
    Nl = Nn // P
    dt = torch.bfloat16
    X32 = torch.randn(B, Nn, Nn, C, device=DEV)
    Go = torch.randn(S, Nn, Nn, device=DEV) / Nn**0.5
    Gd = torch.randn(S, Nn, Nn, device=DEV) / Nn**0.5
    W32 = torch.randn(C * S * S, Hd, device=DEV) / (C * S * S) ** 0.5
    b32 = torch.randn(Hd, device=DEV)
    with torch.no_grad():
        ref = eager.bdgcn_layer_eager(X32, Go, Gd, W32, b32, "relu")
        gop = GraphOperator(Go.to(dt), Gd.to(dt))
        X, W = X32.to(dt), W32.to(dt)
        Vs = [mode1_proj(X[:, :, p * Nl:(p + 1) * Nl, :].contiguous(), W, gop)
              for p in range(P)]
        Vfull = torch.cat(Vs, dim=2)
        Ys = [mode2_bias_act(
                  Vfull[:, p * Nl:(p + 1) * Nl].reshape(B, Nl, Nn, S, Hd)
                  .contiguous(), b32, gop, True)
              for p in range(P)]
        out = torch.cat(Ys, dim=1)
    rel = (out.float() - ref).norm() / ref.norm()
    assert rel < 3e-2, rel.item()
    torch.cuda.synchronize()


def test_fp8_mode1_probe_numerics():
    """fp8 e4m3 mode-1 probe (short-K shape; measurement-only path)."""
    ext = _ext()
    torch.manual_seed(6)
    B, Nn, S, C = 4, 256, 3, 32
    X32 = torch.randn(B, Nn, Nn, C, device=DEV) / Nn**0.25
    G32 = torch.randn(S, Nn, Nn, device=DEV) / Nn**0.25
    ref = torch.einsum("onm,bndl->bmdol", G32, X32)
    GT = G32.transpose(-2, -1).contiguous()
    u8 = ext.bdgcn_mode1_fp8(X32.to(torch.float8_e4m3fn),
                             GT.to(torch.float8_e4m3fn))
    rel = (u8.float() - ref).norm() / ref.norm()
    assert rel < 8e-2, rel.item()


def test_serving_forecaster_gpu():
    """Serving path on GPU: Forecaster rollout through the HIP forward."""
    from mpgcn_amd.data import DataInput
    from mpgcn_amd.models import MPGCN
    from mpgcn_amd.serve import Forecaster

    params = {"synthetic_nodes": 64, "synthetic_days": 60, "seed": 0,
              "split_ratio": [7, 1.5, 1.5], "norm": "none", "hidden_dim": 32,
              "kernel_type": "random_walk_diffusion", "cheby_order": 2,
              "device": DEV, "compute_dtype": "bf16",
              "checkpoint": "/tmp/_serve_ck.pkl"}
    data = DataInput(params=params).load_data()
    model = MPGCN(M=2, K=3, input_dim=1, lstm_hidden_dim=32, lstm_num_layers=1,
                  gcn_hidden_dim=32, gcn_num_layers=3, num_nodes=64)
    torch.save({"epoch": 1, "state_dict": model.state_dict()},
               params["checkpoint"])
    fc = Forecaster(params, data)
    out = fc.forecast(torch.rand(7, 64, 64), dow=3, horizon=4)
    assert out.shape == (4, 64, 64)
    assert torch.isfinite(out).all()
