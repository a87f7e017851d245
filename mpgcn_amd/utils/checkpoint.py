"""Checkpoint I/O.

Two formats:
  * reference-compatible: {'epoch': int, 'state_dict': ...} pickled to
    {output_dir}/{model}_od.pkl (Model_Trainer.py:88,128-129,141) — written by
    the trainer on every validation improvement and at training end; the ONLY
    format test() reads. Round-trips with the reference bit-for-bit (the
    module tree reproduces its state_dict keys, tests/test_model.py).
  * extended (resume): adds optimizer state, RNG states, and the early-stop
    bookkeeping to {output_dir}/{model}_od.resume.pkl so training can resume —
    a capability the reference lacks entirely ("train() never loads it",
    SURVEY.md §5). The compat file stays untouched alongside.
"""

from __future__ import annotations

import os

import torch


def save_compat(path: str, epoch: int, model: torch.nn.Module):
    torch.save({"epoch": epoch, "state_dict": model.state_dict()}, path)


def save_resume(path: str, epoch: int, model, optimizer, val_loss: float,
                patience_count: int):
    torch.save(
        {
            "epoch": epoch,
            "state_dict": model.state_dict(),
            "optimizer": optimizer.state_dict(),
            "val_loss": val_loss,
            "patience_count": patience_count,
            "torch_rng": torch.get_rng_state(),
            "cuda_rng": (
                torch.cuda.get_rng_state_all() if torch.cuda.is_available() else None
            ),
        },
        path,
    )


def load_resume(path: str, model, optimizer):
    """Restores model/optimizer/RNG; returns (next_epoch, val_loss, patience)."""
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    model.load_state_dict(ckpt["state_dict"])
    optimizer.load_state_dict(ckpt["optimizer"])
    torch.set_rng_state(ckpt["torch_rng"])
    if ckpt.get("cuda_rng") is not None and torch.cuda.is_available():
        torch.cuda.set_rng_state_all(ckpt["cuda_rng"])
    return ckpt["epoch"] + 1, ckpt["val_loss"], ckpt["patience_count"]


def resume_path(output_dir: str, model_name: str) -> str:
    return os.path.join(output_dir, f"{model_name}_od.resume.pkl")
