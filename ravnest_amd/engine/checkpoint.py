"""Checkpoint / resume.

The reference only exports final submodels (node.py:692-724 save cascade +
utils.py:232-255 model_fusion) and has NO optimizer-state checkpoint or
mid-training resume. This module adds both, per
stage: model state_dict + optimizer state + engine counters + RNG states,
so an interrupted async-pipeline run resumes exactly.
"""
from __future__ import annotations

from pathlib import Path

import torch


def save_checkpoint(node, path: str | Path | None = None) -> Path:
    """Write a resumable per-stage checkpoint. Call at a drained boundary
    (after wait_for_backwards) so no fpid is in flight."""
    if path is None:
        path = Path(node.template_path or ".") / "checkpoint.pt"
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    state = {
        "model": node.model.state_dict(),
        "optimizer": node.optimizer.state_dict() if node.optimizer else None,
        "n_backwards": node.engine.n_backwards,
        "current_version": node.engine.current_version,
        "forward_pass_id": node.forward_pass_id,
        "latest_backward_id": node.latest_backward_id,
        "rng": {
            "cpu": torch.get_rng_state(),
            "cuda": (torch.cuda.get_rng_state(node.device)
                     if node.device.type == "cuda" else None),
        },
        "rank": node.rank,
        "stage": node.stage,
        "cluster_id": node.cluster_id,
    }
    torch.save(state, path)
    return path


def load_checkpoint(node, path: str | Path | None = None) -> dict:
    if path is None:
        path = Path(node.template_path or ".") / "checkpoint.pt"
    state = torch.load(path, map_location=node.device, weights_only=False)
    node.model.load_state_dict(state["model"])
    if node.optimizer is not None and state["optimizer"] is not None:
        node.optimizer.load_state_dict(state["optimizer"])
    node.engine.n_backwards = state["n_backwards"]
    node.engine.current_version = state["current_version"]
    node.forward_pass_id = state["forward_pass_id"]
    node.latest_backward_id = state["latest_backward_id"]
    torch.set_rng_state(state["rng"]["cpu"])
    if state["rng"]["cuda"] is not None and node.device.type == "cuda":
        torch.cuda.set_rng_state(state["rng"]["cuda"], node.device)
    if node.engine.versioning:
        node.engine.version_to_param.clear()
        node.engine.version_refs.clear()
        node.engine._snapshot_current()
    return state
