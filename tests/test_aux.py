"""Auxiliary subsystems: resume checkpointing, throughput meter, watchdog."""

import re

import pytest
import torch

from tests.test_trainer import _setup


def test_resume_continues_training_fused_adam(tmp_path, capsys):
    # FlatAdam round-trips through the extended resume checkpoint: its
    # state_dict carries the packed flat/exp_avg/exp_avg_sq/step buffers and
    # load restores them with copy_ (checkpoint stored on CPU, optimizer on
    # the training device)
    params, trainer, loaders = _setup(tmp_path, num_epochs=2, resume=True,
                                      optimizer="FusedAdam")
    trainer.train(loaders, ["train", "validate"])
    assert "Epoch 2" in capsys.readouterr().out

    params2, trainer2, loaders2 = _setup(tmp_path, num_epochs=3, resume=True,
                                         optimizer="FusedAdam")
    trainer2.train(loaders2, ["train", "validate"])
    out = capsys.readouterr().out
    assert "resumed from" in out and "Epoch 3" in out
    assert float(trainer2.optimizer.step_t.item()) > 0
    # params still alias the optimizer's flat buffer after resume
    p0 = next(trainer2.model.parameters())
    assert p0.data_ptr() >= trainer2.optimizer.flat.data_ptr()
    assert p0.grad._base.data_ptr() == trainer2.optimizer.flat_grad.data_ptr()


def test_resume_continues_training(tmp_path, capsys):
    params, trainer, loaders = _setup(tmp_path, num_epochs=2, resume=True)
    trainer.train(loaders, ["train", "validate"])
    first = capsys.readouterr().out
    assert "Epoch 2" in first

    # fresh trainer resumes from the extended checkpoint at epoch 3
    params2, trainer2, loaders2 = _setup(tmp_path, num_epochs=4, resume=True)
    trainer2.train(loaders2, ["train", "validate"])
    out = capsys.readouterr().out
    assert "resumed from" in out
    assert "Epoch 3" in out
    assert "Epoch 1," not in out  # did not restart from scratch

    # reference-compatible checkpoint still has the plain schema
    ckpt = torch.load(str(tmp_path) + "/MPGCN_od.pkl", weights_only=False)
    assert set(ckpt.keys()) == {"epoch", "state_dict"}


def test_throughput_meter():
    from mpgcn_amd.utils.profiling import ThroughputMeter

    m = ThroughputMeter(window=4)
    assert m.samples_per_sec == 0.0
    for _ in range(6):
        m.step(10)
    assert m.samples_per_sec > 0


def test_trace_range_inert_without_gpu():
    from mpgcn_amd.utils.profiling import trace_range

    with trace_range("test"):
        pass  # must not raise even without libroctx


def test_rank_watchdog_reraises_single_process():
    from mpgcn_amd.parallel import DistContext, rank_watchdog

    with pytest.raises(ValueError):
        with rank_watchdog(DistContext()):
            raise ValueError("boom")
