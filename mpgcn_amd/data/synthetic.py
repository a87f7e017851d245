"""Synthetic OD-flow data generation.

There is no network access and the reference's private 47-region dataset is not
bundled (Data_Container_OD.py:15-17 hardwires it), so benchmarks and tests run
on synthetic OD tensors with the same statistical shape: non-negative counts
with a weekly (period-7) structure, so the day-of-week dynamic-graph machinery
has real signal to model.
"""

from __future__ import annotations

import torch


def synthetic_od(T: int, N: int, seed: int = 0, device="cpu") -> torch.Tensor:
    """Raw (un-logged) OD counts, shape (T, N, N, 1), float32, >= 0."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    base = torch.rand(N, N, generator=g) * 20.0
    weekly = torch.rand(7, N, N, generator=g) * 10.0
    trend = 1.0 + 0.1 * torch.sin(torch.arange(T, dtype=torch.float32) * (2 * 3.14159 / 90.0))
    noise = torch.randn(T, N, N, generator=g) * 2.0
    days = torch.arange(T) % 7
    flow = base.unsqueeze(0) * trend.view(T, 1, 1) + weekly[days] + noise
    return flow.clamp_min_(0.0).unsqueeze(-1).to(device)


def synthetic_adjacency(N: int, seed: int = 0, degree: int = 8) -> torch.Tensor:
    """Symmetric 0/1 geographic-style adjacency (N, N) with self-loops excluded,
    each node connected to ~`degree` ring neighbors plus random shortcuts."""
    g = torch.Generator(device="cpu").manual_seed(seed + 1)
    A = torch.zeros(N, N)
    idx = torch.arange(N)
    for d in range(1, max(1, degree // 2) + 1):
        A[idx, (idx + d) % N] = 1.0
        A[idx, (idx - d) % N] = 1.0
    shortcuts = torch.rand(N, N, generator=g) < (degree / (4.0 * N))
    A = ((A + shortcuts + shortcuts.T) > 0).float()
    A.fill_diagonal_(0.0)
    return A
