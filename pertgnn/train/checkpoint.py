"""Checkpoint / resume (reference has none — SURVEY.md §5).

Checkpoint = model state_dict (keyed identically to the reference
SAGEDeterministic, including the dead ``edge_linear`` lazy module, quirk 4)
+ optimizer state + RNG states + epoch counter.  Rank 0 writes; all ranks
barrier; every rank loads.
"""
from __future__ import annotations

import os

import torch


def save_checkpoint(path: str, model, optimizer, epoch: int, comm=None, extra: dict | None = None):
    if comm is None or comm.rank == 0:
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        state = {
            "model": model.state_dict(),
            "optimizer": optimizer.state_dict() if optimizer is not None else None,
            "epoch": epoch,
            "rng": {
                "torch": torch.get_rng_state(),
                "cuda": torch.cuda.get_rng_state_all() if torch.cuda.is_available() else None,
            },
            "extra": extra or {},
        }
        tmp = path + ".tmp"
        torch.save(state, tmp)
        os.replace(tmp, path)
    if comm is not None:
        comm.barrier()


def load_checkpoint(path: str, model, optimizer=None, map_location="cpu"):
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None and state.get("optimizer") is not None:
        optimizer.load_state_dict(state["optimizer"])
    rng = state.get("rng", {})
    if rng.get("torch") is not None:
        torch.set_rng_state(rng["torch"].cpu() if torch.is_tensor(rng["torch"]) else rng["torch"])
    if rng.get("cuda") is not None and torch.cuda.is_available():
        try:
            torch.cuda.set_rng_state_all([s.cpu() for s in rng["cuda"]])
        except RuntimeError:
            pass  # different device count than at save time
    return state.get("epoch", 0), state.get("extra", {})
