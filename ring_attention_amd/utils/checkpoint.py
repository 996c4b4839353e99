"""Checkpoint/resume for distributed ring-attention training.

The reference had no persistence beyond raw state_dicts.
This gives resumable training: model + optimizer + step + RNG state, with
rank-0-writes / all-ranks-read semantics and a barrier so no rank resumes
from a half-written file.
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist

from ..parallel import get_rank, is_distributed


def save_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: torch.optim.Optimizer | None = None,
                    step: int = 0, extra: dict | None = None) -> None:
    """Rank 0 writes atomically (tmp + rename); all ranks synchronize after."""
    if get_rank() == 0:
        state = {
            "model": model.state_dict(),
            "optimizer": optimizer.state_dict() if optimizer is not None else None,
            "step": step,
            "torch_rng": torch.get_rng_state(),
            "cuda_rng": torch.cuda.get_rng_state_all() if torch.cuda.is_available() else None,
            "extra": extra or {},
        }
        tmp = path + ".tmp"
        torch.save(state, tmp)
        os.replace(tmp, path)
    if is_distributed():
        dist.barrier()


def load_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: torch.optim.Optimizer | None = None,
                    restore_rng: bool = True,
                    map_location="cpu") -> dict:
    """Every rank loads; returns {'step': int, 'extra': dict}."""
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None and state.get("optimizer") is not None:
        optimizer.load_state_dict(state["optimizer"])
    if restore_rng and state.get("torch_rng") is not None:
        torch.set_rng_state(state["torch_rng"].cpu().to(torch.uint8))
        if torch.cuda.is_available() and state.get("cuda_rng") is not None:
            try:
                torch.cuda.set_rng_state_all([s.cpu().to(torch.uint8) for s in state["cuda_rng"]])
            except RuntimeError:
                pass  # device count mismatch on resume — model/optim state still valid
    if is_distributed():
        dist.barrier()
    return {"step": state.get("step", 0), "extra": state.get("extra", {})}
