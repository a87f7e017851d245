"""GraphOperator operand layouts vs their index definitions (CPU).

The axis kernels consume pre-permuted graph layouts; their packing rules are
documented as index identities (ops/functional.py GoT/A2T/A2/A3T docstrings,
mirrored in ext.hip's binding contracts):

    GoT[.., m, n]        = Go_k[.., n, m]
    A2T[(b,) d, c*Se+s]  = Gd_k[(b,) s, c, d]
    A2 [(b,) c*Se+s, d]  = Gd_k[(b,) s, c, d]
    A3T[(b,) n, o*N+m]   = Go_k[(b,) o, n, m]

with Go_k/Gd_k = the stacks MINUS support 0 in id_first mode. The GPU
equivalence tests validate these transitively through kernel outputs; this
file pins the packing math itself, elementwise, on CPU for static and
dynamic stacks, with and without the identity-first reduction.
"""

import pytest
import torch

from hypothesis import given, settings, strategies as st

from mpgcn_amd.ops import GraphOperator


def _stack(B, S, N, seed, dyn):
    g = torch.Generator().manual_seed(seed)
    shape = (B, S, N, N) if dyn else (S, N, N)
    Go = torch.randn(*shape, generator=g)
    Gd = torch.randn(*shape, generator=g)
    return Go, Gd


@settings(max_examples=30, deadline=None)
@given(S=st.integers(2, 5), N=st.integers(2, 9), B=st.integers(1, 3),
       dyn=st.booleans(), id_first=st.booleans(), seed=st.integers(0, 999))
def test_layout_index_identities(S, N, B, dyn, id_first, seed):
    Go, Gd = _stack(B, S, N, seed, dyn)
    gop = GraphOperator(Go, Gd, id_first=id_first)
    assert gop.id_first == (id_first and S >= 2)
    s0 = 1 if gop.id_first else 0
    Se = S - s0
    go_k = Go[..., s0:, :, :]
    gd_k = Gd[..., s0:, :, :]

    # GoT: transpose of each support matrix
    torch.testing.assert_close(gop.GoT, go_k.transpose(-2, -1), rtol=0, atol=0)

    A2T, A2, A3T = gop.A2T, gop.A2, gop.A3T
    if not dyn:
        A2T, A2, A3T = A2T.unsqueeze(0), A2.unsqueeze(0), A3T.unsqueeze(0)
        go_k, gd_k = go_k.unsqueeze(0), gd_k.unsqueeze(0)
    assert A2T.shape == (B if dyn else 1, N, N * Se)
    assert A2.shape == (B if dyn else 1, N * Se, N)
    assert A3T.shape == (B if dyn else 1, N, Se * N)
    for b in range(B if dyn else 1):
        for s in range(Se):
            for c in range(N):
                torch.testing.assert_close(
                    A2T[b, :, c * Se + s], gd_k[b, s, c, :], rtol=0, atol=0)
                torch.testing.assert_close(
                    A2[b, c * Se + s, :], gd_k[b, s, c, :], rtol=0, atol=0)
            torch.testing.assert_close(
                A3T[b, :, s * N:(s + 1) * N], go_k[b, s, :, :], rtol=0, atol=0)


def test_layouts_cached_per_operator():
    Go, Gd = _stack(1, 3, 6, 0, False)
    gop = GraphOperator(Go, Gd, id_first=False)
    assert gop.A2T is gop.A2T  # lazy, computed once
    assert gop.GoT is gop.GoT
    assert gop.A2 is gop.A2 and gop.A3T is gop.A3T


def test_id_first_requires_both_tags():
    Go, Gd = _stack(1, 3, 6, 1, False)
    Go._identity_first = True  # only one operand tagged
    gop = GraphOperator(Go, Gd)
    assert not gop.id_first
    Gd._identity_first = True
    gop2 = GraphOperator(Go, Gd)
    assert gop2.id_first
