"""Dynamic (day-of-week) OD-correlation graph construction.

Capability of the reference's ``DataInput.construct_dyn_G`` (Data_Container_OD.py:39-59):
for each phase t of a perceived period (default 7, day-of-week), average the training
history's OD matrices at that phase, then build

  O_G_t[i, j] = cosine_distance(row_i, row_j)   (origin-similarity graph,   paper eq. 6)
  D_G_t[i, j] = cosine_distance(col_i, col_j)   (destination-similarity,    paper eq. 7)

The reference does this with an O(period * N^3) doubly-nested Python loop of scipy
calls — intractable beyond a few hundred regions. Here each phase is
two normalized GEMMs (R @ R^T), batched over all phases in one shot and runnable on
device; at N = 4096 this is a pair of (7, 4096, 4096) batched matmuls.

Deliberately NOT replicated (documented quirk): the reference's D-graph
mixes column i with ROW j (``distance.cosine(OD_t_avg[:, i], OD_t_avg[j, :])``,
Data_Container_OD.py:56); we compute the symmetric column-column distance the paper
describes. Zero rows/columns get cosine distance 0 to themselves and 1 to anything
else (scipy would emit NaN there).
"""

from __future__ import annotations

import torch


def _cosine_distance_gram(rows: torch.Tensor, eps: float = 1e-12) -> torch.Tensor:
    """(P, N, F) row sets -> (P, N, N) pairwise cosine DISTANCE (1 - similarity)."""
    norms = rows.norm(dim=-1, keepdim=True)
    unit = rows / norms.clamp(min=eps)
    sim = torch.bmm(unit, unit.transpose(-2, -1))
    dist = 1.0 - sim
    # Zero vectors: unit row is 0 -> sim 0 -> dist 1 off-diagonal; force the diagonal
    # to exactly 0 (self-distance), which also kills fp roundoff on the diagonal.
    P, N, _ = dist.shape
    idx = torch.arange(N, device=dist.device)
    dist[:, idx, idx] = 0.0
    return dist


def _ref_quirk_distance(cols: torch.Tensor, rows: torch.Tensor,
                        eps: float = 1e-12) -> torch.Tensor:
    """Reference-exact D-graph: D[i, j] = cosine_distance(col_i, ROW_j) —
    the column/row mixing of Data_Container_OD.py:56, reproduced verbatim for
    bit-parity validation runs (no diagonal forcing: cosine(col_i, row_i) is
    generally nonzero). cols/rows: (P, N, F) with cols[p, i] = OD_avg[:, i]
    and rows[p, j] = OD_avg[j, :]."""
    cu = cols / cols.norm(dim=-1, keepdim=True).clamp(min=eps)
    ru = rows / rows.norm(dim=-1, keepdim=True).clamp(min=eps)
    return 1.0 - torch.bmm(cu, ru.transpose(-2, -1))


def construct_dynamic_graphs(
    OD_history: torch.Tensor,
    period: int = 7,
    ref_quirks: bool = False,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Period-phase averaged OD -> (O_dyn_G, D_dyn_G), each (N, N, period).

    OD_history: (T, N, N) or (T, N, N, 1) UN-normalized flow, already truncated to the
    training split and to a whole number of periods (caller's job, matching
    Data_Container_OD.py:40-42 which dumps the remainder weeks).
    Output layout matches the reference's (N, N, period) stacking
    (Data_Container_OD.py:59) so downstream day-of-week indexing is identical.

    ref_quirks=True reproduces the reference's D-graph computation EXACTLY
    (Data_Container_OD.py:53-56): column i against ROW j, no symmetric
    correction and no diagonal forcing — for users validating bit-parity
    against the upstream framework on the same real dataset.
    """
    if OD_history.dim() == 4:
        OD_history = OD_history.squeeze(-1)
    T, N, _ = OD_history.shape
    if T % period != 0:
        raise ValueError(
            f"OD_history length {T} is not a whole number of periods ({period}); "
            "truncate to full periods first (the reference dumps the remainder)."
        )
    # (T, N, N) -> (weeks, period, N, N) -> phase means (period, N, N)
    phase_avg = OD_history.reshape(T // period, period, N, N).mean(dim=0)

    if ref_quirks:
        # diagonal NOT forced to zero in quirk mode (scipy's cosine leaves
        # ~1e-16 roundoff there; the reference keeps whatever scipy returns)
        ru = phase_avg / phase_avg.norm(dim=-1, keepdim=True).clamp(min=1e-12)
        O_dyn = 1.0 - torch.bmm(ru, ru.transpose(-2, -1))
        D_dyn = _ref_quirk_distance(phase_avg.transpose(-2, -1), phase_avg)
    else:
        O_dyn = _cosine_distance_gram(phase_avg)  # rows = origin profiles
        D_dyn = _cosine_distance_gram(phase_avg.transpose(-2, -1))  # rows = dest profiles

    # (period, N, N) -> (N, N, period), the reference's stacking layout
    return O_dyn.permute(1, 2, 0).contiguous(), D_dyn.permute(1, 2, 0).contiguous()
