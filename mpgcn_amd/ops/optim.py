"""FlatAdam: fused single-buffer Adam for the MI355X training step.

torch.optim.Adam in capturable mode (required under hipGraph capture) issues
~45 tiny kernels per step for this model family — per-tensor foreach slices,
device-side bias-correction pow/div chains, and per-tensor grad zeroing fills.
Those launches survive graph replay as real kernel time. FlatAdam packs every
f32 parameter into ONE contiguous buffer (parameters become views into it),
pins the matching gradient views so autograd accumulates in place, and runs
the whole update as a single vectorized HIP kernel plus a one-thread step
bump (ops/hip/elemwise.hip adam_flat_kernel). zero_grad() is one fill.

Math matches torch.optim.Adam defaults exactly (L2-style weight decay, not
AdamW; no amsgrad): reference Model_Trainer.py:60-63 trains with plain Adam.
The device-resident step counter keeps the whole schedule capture-safe.

CPU fallback: the same flat-buffer update in stock torch ops, so the class is
usable (and testable) without a GPU.
"""

from __future__ import annotations

from typing import Iterable

import torch


class FlatAdam:
    """Drop-in Adam for f32 master weights, fused into one kernel per step.

    Constraints vs torch.optim.Adam (asserted, not silently ignored):
      * all parameters must be f32 and on one device;
      * parameters are REPOINTED to views of an internal flat buffer, and
        ``p.grad`` is pre-assigned a view of a flat gradient buffer — construct
        the optimizer after ``model.to(device)`` and before taking any other
        reference to ``p.data``;
      * ``zero_grad`` always keeps the (stable) buffers, as hipGraph capture
        requires — ``set_to_none`` is accepted and ignored.
    """

    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-3,
                 betas: tuple[float, float] = (0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        plist = [p for p in params if p.requires_grad]
        if not plist:
            raise ValueError("FlatAdam: no trainable parameters")
        device = plist[0].device
        for p in plist:
            if p.dtype != torch.float32 or p.device != device:
                raise ValueError(
                    "FlatAdam requires f32 parameters on a single device; got "
                    f"{p.dtype} on {p.device}")
        total = sum(p.numel() for p in plist)
        self.flat = torch.empty(total, dtype=torch.float32, device=device)
        self.flat_grad = torch.zeros_like(self.flat)
        off = 0
        for p in plist:
            n = p.numel()
            self.flat[off:off + n] = p.detach().reshape(-1)
            p.data = self.flat[off:off + n].view(p.shape)
            p.grad = self.flat_grad[off:off + n].view(p.shape)
            off += n
        self.exp_avg = torch.zeros_like(self.flat)
        self.exp_avg_sq = torch.zeros_like(self.flat)
        self.step_t = torch.zeros(1, dtype=torch.float32, device=device)
        self.lr, self.betas, self.eps = lr, betas, eps
        self.weight_decay = weight_decay
        # minimal torch-optimizer surface for callers that poke at groups
        self.param_groups = [{
            "params": plist, "lr": lr, "betas": betas, "eps": eps,
            "weight_decay": weight_decay,
        }]

    def zero_grad(self, set_to_none: bool = True) -> None:  # noqa: ARG002
        self.flat_grad.zero_()

    @torch.no_grad()
    def step(self) -> None:
        lr = self.param_groups[0]["lr"]  # honor external lr edits
        b1, b2 = self.betas
        if self.flat.is_cuda:
            from mpgcn_amd import ops as _ops

            _ops.get_ext().adam_flat(self.flat, self.flat_grad, self.exp_avg,
                                     self.exp_avg_sq, self.step_t, lr, b1, b2,
                                     self.eps, self.weight_decay)
            return
        self.step_t += 1.0
        t = float(self.step_t.item())
        g = self.flat_grad
        if self.weight_decay:
            g = g + self.weight_decay * self.flat
        self.exp_avg.mul_(b1).add_(g, alpha=1.0 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=1.0 - b2)
        bc1 = 1.0 - b1 ** t
        bc2 = 1.0 - b2 ** t
        denom = self.exp_avg_sq.sqrt().div_(bc2 ** 0.5).add_(self.eps)
        self.flat.addcdiv_(self.exp_avg, denom, value=-lr / bc1)

    def state_dict(self) -> dict:
        return {
            "flat": self.flat, "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq, "step_t": self.step_t,
            "lr": self.param_groups[0]["lr"], "betas": self.betas,
            "eps": self.eps, "weight_decay": self.weight_decay,
        }

    def load_state_dict(self, state: dict) -> None:
        self.flat.copy_(state["flat"])
        self.exp_avg.copy_(state["exp_avg"])
        self.exp_avg_sq.copy_(state["exp_avg_sq"])
        self.step_t.copy_(state["step_t"])
        self.param_groups[0]["lr"] = state["lr"]
        self.betas, self.eps = state["betas"], state["eps"]
        self.weight_decay = state["weight_decay"]
