"""Graph-support builders: raw flow/adjacency matrices -> K support matrices.

Capability parity with the reference's ``Adj_Processor`` (GCN.py:49-138): the same
four kernel types with the same support-count contract (Model_Trainer.py:24-36), but
fully batched and device-resident — the reference loops over the batch in Python on
the CPU twice per training step (GCN.py:64-98, Model_Trainer.py:106); here every
builder is a handful of batched tensor ops that run on-GPU, so per-step dynamic
support construction costs no host<->device round trip.

Kernel types (support count K_s):
  localpool                  (Kipf ICLR'17)      K_s = 1
  chebyshev                  (Defferrard NIPS'16) K_s = order + 1
  random_walk_diffusion      (Li ICLR'18, DCRNN) K_s = order + 1
  dual_random_walk_diffusion                     K_s = 2*order + 1

Notes vs the reference, deliberate (SURVEY.md "quirks"):
  * The reference's ``rescale_laplacian`` calls ``torch.eig`` (removed from modern
    torch) inside a bare ``except`` that silently falls back to lambda_max = 2
    (GCN.py:116-126) — on every modern install the fallback ALWAYS fires. We make
    that behavior explicit: ``lambda_max=2.0`` by default, or ``lambda_max=None`` to
    compute it per-matrix with a power iteration (batched, on device).
  * ``symmetric_normalize`` guards empty rows (inf -> 0) the same way the reference
    guards ``random_walk_normalize`` (GCN.py:105); the reference leaves the symmetric
    path unguarded and produces inf on empty rows.
"""

from __future__ import annotations

import torch

KERNEL_TYPES = (
    "chebyshev",
    "localpool",
    "random_walk_diffusion",
    "dual_random_walk_diffusion",
)


def tag_like(dst: torch.Tensor, src: torch.Tensor) -> torch.Tensor:
    """Propagate the _identity_first tag through views/squeezes/casts (tensor
    attributes do not survive torch ops)."""
    if getattr(src, "_identity_first", False):
        dst._identity_first = True
    return dst


def get_support_K(kernel_type: str, cheby_order: int) -> int:
    """Support count per kernel type. Contract of Model_Trainer.py:24-36."""
    if kernel_type == "localpool":
        if cheby_order != 1:
            raise ValueError("localpool requires cheby_order == 1")
        return 1
    if kernel_type in ("chebyshev", "random_walk_diffusion"):
        return cheby_order + 1
    if kernel_type == "dual_random_walk_diffusion":
        return 2 * cheby_order + 1
    raise ValueError(
        f"Invalid kernel_type {kernel_type!r}. Must be one of {list(KERNEL_TYPES)}."
    )


def random_walk_normalize(A: torch.Tensor) -> torch.Tensor:
    """P = D^-1 A, batched over leading dims. Empty rows -> zero rows (GCN.py:102-108)."""
    d = A.sum(dim=-1)
    d_inv = torch.where(d == 0, torch.zeros_like(d), d.reciprocal())
    return d_inv.unsqueeze(-1) * A


def symmetric_normalize(A: torch.Tensor) -> torch.Tensor:
    """D^-1/2 A D^-1/2, batched. Empty rows guarded (cf. GCN.py:110-114)."""
    d = A.sum(dim=-1)
    d_inv_sqrt = torch.where(d == 0, torch.zeros_like(d), d.rsqrt())
    return d_inv_sqrt.unsqueeze(-1) * A * d_inv_sqrt.unsqueeze(-2)


def _batched_eye(B: int, N: int, *, dtype, device) -> torch.Tensor:
    return torch.eye(N, dtype=dtype, device=device).expand(B, N, N)


def power_iteration_lambda_max(
    M: torch.Tensor, iters: int = 50, eps: float = 1e-12
) -> torch.Tensor:
    """Largest-|eigenvalue| estimate per batch matrix, on device. (B, N, N) -> (B,)."""
    B, N, _ = M.shape
    v = torch.ones(B, N, 1, dtype=M.dtype, device=M.device) / (N**0.5)
    for _ in range(iters):
        v = torch.bmm(M, v)
        v = v / (v.norm(dim=1, keepdim=True) + eps)
    Mv = torch.bmm(M, v)
    return (v * Mv).sum(dim=(1, 2))


def chebyshev_polynomials(x: torch.Tensor, order: int) -> torch.Tensor:
    """[T_0 = I, T_1 = x, T_k = 2 x T_{k-1} - T_{k-2}] stacked on dim 1.

    x: (B, N, N)  ->  (B, order+1, N, N). Matches GCN.py:128-138 (note the reference
    multiplies x @ T_{k-1}, i.e. left-multiplication by x).
    """
    B, N, _ = x.shape
    terms = [_batched_eye(B, N, dtype=x.dtype, device=x.device)]
    if order >= 1:
        terms.append(x)
    for k in range(2, order + 1):
        # 2 * x @ T_{k-1} - T_{k-2}, one fused batched GEMM.
        terms.append(torch.baddbmm(terms[k - 2], x, terms[k - 1], beta=-1.0, alpha=2.0))
    return torch.stack(terms, dim=1)


def build_supports(
    flow: torch.Tensor,
    kernel_type: str,
    cheby_order: int,
    lambda_max: float | None = 2.0,
) -> torch.Tensor:
    """Raw flow/adjacency (B, N, N) -> support stack (B, K_s, N, N), batched on device.

    Semantics of GCN.py:56-100 without the per-sample Python loop:
      localpool:                I + sym_norm(A)
      chebyshev:                T_k(rescale(I - sym_norm(A)))            k = 0..order
      random_walk_diffusion:    T_k(P_fwd^T), P_fwd = D^-1 A             k = 0..order
      dual_random_walk_diffusion: fwd series ++ bwd series[1:] (shared T_0 = I)
    """
    if flow.dim() != 3:
        raise ValueError(f"flow must be (B, N, N); got {tuple(flow.shape)}")
    K = get_support_K(kernel_type, cheby_order)

    if flow.is_cuda and (
            (cheby_order >= 1 and (
                kernel_type == "random_walk_diffusion"
                or kernel_type == "dual_random_walk_diffusion"
                or (kernel_type == "chebyshev" and lambda_max is not None)))
            or kernel_type == "localpool"):
        # fused HIP builds (K8): a handful of launches instead of the ~10+
        # stock-op chain below; the torch path remains the numerics oracle
        # (tests/test_supports.py) and the CPU / localpool / power-iteration
        # fallback
        from mpgcn_amd import ops as _ops

        ext = _ops.get_ext()
        f32 = flow.float().contiguous()
        if kernel_type == "localpool":
            return ext.localpool_supports(f32)  # single support: no id tag
        if kernel_type == "random_walk_diffusion":
            out = ext.rwd_supports(f32, cheby_order)
        elif kernel_type == "dual_random_walk_diffusion":
            out = ext.dual_rwd_supports(f32, cheby_order)
        else:
            out = ext.cheby_supports(f32, cheby_order, float(lambda_max))
        out._identity_first = True
        return out

    if kernel_type == "localpool":
        sup = _batched_eye(
            flow.shape[0], flow.shape[1], dtype=flow.dtype, device=flow.device
        ) + symmetric_normalize(flow)
        out = sup.unsqueeze(1)
    elif kernel_type == "chebyshev":
        L = _batched_eye(
            flow.shape[0], flow.shape[1], dtype=flow.dtype, device=flow.device
        ) - symmetric_normalize(flow)
        if lambda_max is None:
            lam = power_iteration_lambda_max(L).clamp(min=1e-6).view(-1, 1, 1)
        else:
            lam = torch.as_tensor(lambda_max, dtype=L.dtype, device=L.device)
        L_rescaled = (2.0 / lam) * L - _batched_eye(
            L.shape[0], L.shape[1], dtype=L.dtype, device=L.device
        )
        out = chebyshev_polynomials(L_rescaled, cheby_order)
    elif kernel_type == "random_walk_diffusion":
        P_fwd = random_walk_normalize(flow)
        out = chebyshev_polynomials(P_fwd.transpose(-2, -1), cheby_order)
    elif kernel_type == "dual_random_walk_diffusion":
        P_fwd = random_walk_normalize(flow)
        P_bwd = random_walk_normalize(flow.transpose(-2, -1))
        fwd = chebyshev_polynomials(P_fwd.transpose(-2, -1), cheby_order)
        bwd = chebyshev_polynomials(P_bwd.transpose(-2, -1), cheby_order)
        out = torch.cat([fwd, bwd[:, 1:]], dim=1)  # shared order-0 term I
    else:
        raise ValueError(
            f"Invalid kernel_type {kernel_type!r}. Must be one of {list(KERNEL_TYPES)}."
        )

    assert out.shape[1] == K, (out.shape, K)
    if kernel_type != "localpool" and K >= 2:
        # every Chebyshev-family stack starts with T_0 = I by construction —
        # the GPU layer skips the identity support's products entirely
        # (ops/functional.py GraphOperator.id_first)
        out._identity_first = True
    return out
