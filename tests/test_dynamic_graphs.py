"""GPU-style batched cosine dynamic-graph builder vs scipy reference
(Data_Container_OD.py:39-59 semantics, with the documented D-graph fix)."""

import numpy as np
import torch
from scipy.spatial import distance

from mpgcn_amd.graph import construct_dynamic_graphs


def _scipy_reference(OD_history, period=7):
    T, N, _ = OD_history.shape
    O_list, D_list = [], []
    for t in range(period):
        avg = OD_history[t::period].mean(axis=0)
        O_G = np.zeros((N, N))
        D_G = np.zeros((N, N))
        for i in range(N):
            for j in range(N):
                O_G[i, j] = distance.cosine(avg[i, :], avg[j, :])
                # column-column distance (the reference's row/col mix at
                # Data_Container_OD.py:56 is a documented quirk, not replicated)
                D_G[i, j] = distance.cosine(avg[:, i], avg[:, j])
        O_list.append(O_G)
        D_list.append(D_G)
    return np.stack(O_list, axis=-1), np.stack(D_list, axis=-1)


def test_matches_scipy():
    rng = np.random.default_rng(0)
    T, N = 28, 12
    hist = rng.uniform(0.1, 5.0, size=(T, N, N)).astype(np.float64)
    O_ref, D_ref = _scipy_reference(hist)
    O_out, D_out = construct_dynamic_graphs(torch.from_numpy(hist))
    assert O_out.shape == (N, N, 7) and D_out.shape == (N, N, 7)
    np.testing.assert_allclose(O_out.numpy(), O_ref, atol=1e-6)
    np.testing.assert_allclose(D_out.numpy(), D_ref, atol=1e-6)


def test_channel_axis_and_truncation_contract():
    T, N = 21, 5
    hist = torch.rand(T, N, N, 1) + 0.1
    O1, D1 = construct_dynamic_graphs(hist)
    O2, D2 = construct_dynamic_graphs(hist.squeeze(-1))
    assert torch.equal(O1, O2) and torch.equal(D1, D2)
    try:
        construct_dynamic_graphs(torch.rand(20, N, N))
        raise AssertionError("partial periods must be rejected")
    except ValueError:
        pass
