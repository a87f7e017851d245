"""FlatAdam (fused flat-buffer Adam, ops/optim.py) vs torch.optim.Adam.

The optimizer repoints parameters and gradients into packed flat buffers, so
besides step-for-step numerical parity we verify the structural contract:
autograd accumulates IN PLACE into the preset grad views (the flat gradient
buffer sees every backward), and DDP-style in-place grad edits reach the
buffer too.
"""

import copy

import pytest
import torch

from mpgcn_amd.ops.optim import FlatAdam


def _tiny_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(6, 16), torch.nn.ReLU(), torch.nn.Linear(16, 3)
    )


def _train(model, opt, steps, device="cpu"):
    torch.manual_seed(123)
    for _ in range(steps):
        x = torch.randn(8, 6, device=device)
        y = torch.randn(8, 3, device=device)
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
    return loss


@pytest.mark.parametrize("wd", [0.0, 0.01])
def test_flat_adam_matches_torch_adam_cpu(wd):
    m_ref = _tiny_model()
    m_flat = copy.deepcopy(m_ref)
    opt_ref = torch.optim.Adam(m_ref.parameters(), lr=3e-3, weight_decay=wd)
    opt_flat = FlatAdam(m_flat.parameters(), lr=3e-3, weight_decay=wd)
    _train(m_ref, opt_ref, 6)
    _train(m_flat, opt_flat, 6)
    for pr, pf in zip(m_ref.parameters(), m_flat.parameters()):
        torch.testing.assert_close(pf, pr, rtol=1e-5, atol=1e-6)


def test_flat_adam_grad_views_accumulate_in_place():
    m = _tiny_model()
    opt = FlatAdam(m.parameters(), lr=1e-3)
    x = torch.randn(4, 6)
    m(x).sum().backward()
    m(x).sum().backward()  # second backward must accumulate, not replace
    p0 = next(m.parameters())
    assert p0.grad._base is not None and p0.grad._base.data_ptr() == \
        opt.flat_grad.data_ptr(), "autograd replaced the preset grad view"
    assert opt.flat_grad.abs().sum() > 0
    # DDP finalize writes via g.copy_(...): must reach the flat buffer
    p0.grad.copy_(torch.full_like(p0.grad, 7.0))
    assert (opt.flat_grad[: p0.numel()] == 7.0).all()
    opt.zero_grad()
    assert opt.flat_grad.abs().sum() == 0 and p0.grad.abs().sum() == 0


def test_flat_adam_state_roundtrip():
    m = _tiny_model()
    opt = FlatAdam(m.parameters(), lr=2e-3)
    _train(m, opt, 3)
    state = {k: (v.clone() if torch.is_tensor(v) else v)
             for k, v in opt.state_dict().items()}
    m2 = _tiny_model(seed=1)
    opt2 = FlatAdam(m2.parameters(), lr=9.0)
    opt2.load_state_dict(state)
    for pa, pb in zip(m.parameters(), m2.parameters()):
        torch.testing.assert_close(pa, pb)
    la = _train(m, opt, 2)
    lb = _train(m2, opt2, 2)
    torch.testing.assert_close(la, lb)


def test_flat_adam_rejects_mixed_dtype():
    m = _tiny_model().to(torch.float64)
    with pytest.raises(ValueError):
        FlatAdam(m.parameters())


@pytest.mark.gpu
@pytest.mark.parametrize("wd", [0.0, 0.01])
def test_flat_adam_hip_kernel_matches_torch_adam(wd):
    dev = "cuda"
    m_ref = _tiny_model().to(dev)
    m_flat = copy.deepcopy(m_ref)
    opt_ref = torch.optim.Adam(m_ref.parameters(), lr=3e-3, weight_decay=wd)
    opt_flat = FlatAdam(m_flat.parameters(), lr=3e-3, weight_decay=wd)
    _train(m_ref, opt_ref, 6, device=dev)
    _train(m_flat, opt_flat, 6, device=dev)
    for pr, pf in zip(m_ref.parameters(), m_flat.parameters()):
        torch.testing.assert_close(pf, pr, rtol=1e-4, atol=1e-6)
    assert float(opt_flat.step_t.item()) == 6.0


@pytest.mark.gpu
def test_flat_adam_tail_elements():
    # non-multiple-of-4 flat size exercises the scalar tail path
    dev = "cuda"
    p_ref = torch.randn(7, 3, device=dev)  # 21 elements
    p_a = torch.nn.Parameter(p_ref.clone())
    p_b = torch.nn.Parameter(p_ref.clone())
    opt_a = torch.optim.Adam([p_a], lr=1e-2)
    opt_b = FlatAdam([p_b], lr=1e-2)
    torch.manual_seed(7)
    for _ in range(4):
        g = torch.randn(7, 3, device=dev)
        opt_a.zero_grad(); opt_b.zero_grad()
        p_a.grad = g.clone()
        p_b.grad.copy_(g)
        opt_a.step(); opt_b.step()
    torch.testing.assert_close(p_b, p_a, rtol=1e-4, atol=1e-6)
