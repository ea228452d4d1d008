"""Shared helpers: determinism, config IO, dtype codes, model fusion.

Capability parity with the reference's ravnest/utils.py (set_seed :196,
load_node_json_configs :139, model_fusion :232, fp16 compression :184).
The pickle-chunk generators and gRPC framing of the reference are gone:
tensors move device-to-device over RCCL (see ravnest_amd/comm/), so the
only serialization left is the artifact IO of the planner.
"""
from __future__ import annotations

import json
import os
import random
from pathlib import Path

import numpy as np
import torch

# ---------------------------------------------------------------------------
# Wire dtype codes (shared by comm headers). bf16 is the preferred pipeline
# dtype on MI355X; fp32 kept for CPU tests and reference numerics checks.
# ---------------------------------------------------------------------------
DTYPE_CODES = {
    torch.float32: 0,
    torch.float64: 1,
    torch.float16: 2,
    torch.bfloat16: 3,
    torch.int64: 4,
    torch.int32: 5,
    torch.bool: 6,
    torch.uint8: 7,
    torch.int16: 8,
    torch.int8: 9,
}
CODE_DTYPES = {v: k for k, v in DTYPE_CODES.items()}


_TRACE = os.environ.get("RAVNEST_TRACE", "0") == "1"


class trace_range:
    """rocTX range (via torch's nvtx shim, which maps to roctx on ROCm).
    Enabled with RAVNEST_TRACE=1; shows per-stage/microbatch phases in
    rocprofv3 runtime traces (SURVEY.md section 5 tracing parity)."""

    def __init__(self, name: str):
        self.name = name
        self.active = _TRACE and torch.cuda.is_available()

    def __enter__(self):
        if self.active:
            torch.cuda.nvtx.range_push(self.name)
        return self

    def __exit__(self, *a):
        if self.active:
            torch.cuda.nvtx.range_pop()
        return False


def set_seed(seed: int) -> None:
    """Deterministic seeding across python/numpy/torch/ROCm.

    Parity: reference ravnest/utils.py:196-209 (set_seed). The reference's
    data-order synchronization across DP replicas depends on this.
    """
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():  # ROCm path: torch.cuda IS the HIP device
        torch.cuda.manual_seed_all(seed)
    os.environ["PYTHONHASHSEED"] = str(seed)
    torch.backends.cudnn.deterministic = True  # MIOpen determinism on ROCm
    torch.backends.cudnn.benchmark = False


def gpu_usage(device=None) -> dict:
    """GPU + host memory telemetry (parity: reference utils.py:211-221
    check_gpu_usage via NVML, called around every fwd/bwd at
    node.py:447-450 / compute.py:148-304; host RAM at node.py:490,554).

    MI355X-native: torch.cuda.mem_get_info goes through the ROCm runtime
    (no NVML on AMD), plus the allocator's own view of reserved/allocated
    bytes. Returns {} fields gracefully on CPU-only hosts.
    """
    out: dict = {}
    if torch.cuda.is_available():
        dev = device if device is not None else torch.cuda.current_device()
        free_b, total_b = torch.cuda.mem_get_info(dev)
        out.update(
            gpu_free_mb=free_b // 2**20,
            gpu_total_mb=total_b // 2**20,
            gpu_used_mb=(total_b - free_b) // 2**20,
            torch_allocated_mb=torch.cuda.memory_allocated(dev) // 2**20,
            torch_reserved_mb=torch.cuda.memory_reserved(dev) // 2**20,
        )
    try:
        import psutil
        vm = psutil.virtual_memory()
        out.update(host_ram_percent=vm.percent,
                   host_ram_used_mb=vm.used // 2**20)
    except Exception:
        pass
    return out


def wire_cast(t: torch.Tensor, wire_dtype: torch.dtype | None) -> torch.Tensor:
    """Optional lossy on-the-wire cast (parity: fp16 compression,
    reference ravnest/utils.py:184-194). On MI355X the natural wire dtype
    is bf16 (no clamping needed: bf16 keeps fp32 range)."""
    if wire_dtype is None or t.dtype == wire_dtype:
        return t
    if t.is_floating_point():
        return t.to(wire_dtype)
    return t


def load_node_json_configs(name: str, base_dir: str = "node_data") -> dict:
    """Load node_data/nodes/<name>.json (parity: reference utils.py:139-155)."""
    path = Path(base_dir) / "nodes" / f"{name}.json"
    with open(path) as f:
        return json.load(f)


def model_fusion(cluster_id: int = 0, base_dir: str = "node_data",
                 out_dir: str = "trained") -> dict:
    """Merge every stage's trained state_dict back into one full-model
    state_dict (parity: reference utils.py:232-255).

    Stage artifacts store parameters under their fx-submodule names; the
    planner records the stage->original-name mapping in `param_names.json`
    so fusion restores the original module names exactly.
    """
    cluster_dir = Path(base_dir) / f"cluster_{cluster_id}"
    fused: dict = {}
    stage_dirs = sorted(cluster_dir.glob("stage_*"))
    if not stage_dirs:
        raise FileNotFoundError(f"no stage artifacts under {cluster_dir}")
    for stage_dir in stage_dirs:
        sd_path = stage_dir / "trained_state_dict.pt"
        if not sd_path.exists():
            sd_path = stage_dir / "state_dict.pt"
        sd = torch.load(sd_path, map_location="cpu", weights_only=True)
        name_map_path = stage_dir / "param_names.json"
        if name_map_path.exists():
            with open(name_map_path) as f:
                name_map = json.load(f)
        else:
            name_map = {}
        for k, v in sd.items():
            fused[name_map.get(k, k)] = v
    out = Path(out_dir)
    out.mkdir(parents=True, exist_ok=True)
    torch.save(fused, out / "trained_state_dict.pt")
    return fused


def current_rng_states(device: torch.device) -> dict:
    """Snapshot CPU (+device) RNG state for versioned recompute replay
    (parity: reference compute.py:63-68)."""
    states = {"cpu": torch.get_rng_state()}
    if device.type == "cuda" and torch.cuda.is_available():
        states["cuda"] = torch.cuda.get_rng_state(device)
    return states


def restore_rng_states(states: dict, device: torch.device) -> None:
    torch.set_rng_state(states["cpu"])
    if "cuda" in states and device.type == "cuda" and torch.cuda.is_available():
        torch.cuda.set_rng_state(states["cuda"], device)
