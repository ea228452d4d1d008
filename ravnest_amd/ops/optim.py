"""Fused multi-tensor optimizers: Adam, SGD+momentum, LAMB.

Reference workload parity: Adam (CNN), SGD+momentum+wd (ResNet/Inception),
LAMB (BERT, torch_optimizer.Lamb) — SURVEY.md section 2.3 optimizer row.
GPU path: ONE kernel launch per step applies the update across every
parameter chunk (csrc/optim.hip, apex-style chunked multi-tensor apply);
LAMB adds a norm-reduction phase for the per-tensor trust ratio. CPU
path: plain torch math (used by the CPU pipeline tests).
"""
from __future__ import annotations

import torch
from torch.optim import Optimizer

from ._ext import get_ext


def _on_gpu(params):
    return any(p.is_cuda for p in params)


class FusedAdam(Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._graph = None  # hipGraph-capture buffers (see graph_step)

    # ---- hipGraph capture support (engine/graphstep.py) --------------
    # lr and the bias-correction step live in DEVICE buffers so graph
    # replays see live values; the chunk table is built once at capture
    # (param/grad/state pointers are stable across replays).
    def enable_graph_capture(self, device) -> bool:
        if len(self.param_groups) != 1:
            return False
        self._ensure_state()
        start = 0
        for st in self.state.values():
            if "step" in st:
                start = max(start, st["step"])
        self._graph = {
            "lr": torch.zeros(1, dtype=torch.float32, device=device),
            "step": torch.full((1,), start, dtype=torch.int64,
                               device=device),
            "table": None,
        }
        self.sync_lr()
        return True

    def _ensure_state(self):
        for group in self.param_groups:
            for p in group["params"]:
                st = self.state[p]
                if len(st) == 0:
                    st["step"] = 0
                    st["m"] = torch.zeros_like(p, dtype=torch.float32)
                    st["v"] = torch.zeros_like(p, dtype=torch.float32)
                    if p.dtype == torch.bfloat16 and p.is_cuda:
                        st["master"] = p.detach().float().clone()

    def sync_lr(self):
        """Host -> device lr refresh; call OUTSIDE the graph, before each
        replay (lets lr schedules work under capture)."""
        self._graph["lr"].fill_(float(self.param_groups[0]["lr"]))

    def bump_host_steps(self):
        """Mirror the device step counter into the python state (kept
        consistent for checkpoints); call once per replayed step."""
        for st in self.state.values():
            if "step" in st:
                st["step"] += 1

    @torch.no_grad()
    def build_graph_table(self):
        """Allocate stable grad buffers and prebuild the device chunk
        table. Must run OUTSIDE capture (the table upload is a blocking
        H2D copy); the captured step then zeroes these buffers, lets
        backward ACCUMULATE into them, and applies the one-kernel update.
        Returns the stable grad list (for the capture's zero pass)."""
        ext = get_ext(required=True)
        group = self.param_groups[0]
        grads = []
        ps = []
        for p in group["params"]:
            if not p.requires_grad:
                continue
            p.grad = torch.zeros_like(p)
            ps.append(p)
            grads.append(p.grad)
        ms = [self.state[p]["m"] for p in ps]
        vs = [self.state[p]["v"] for p in ps]
        masters = ([self.state[p]["master"] for p in ps]
                   if ps and ps[0].dtype == torch.bfloat16 else [])
        self._graph["table"] = ext.adam_build_table(ps, grads, ms, vs,
                                                    masters)
        self._graph["grads"] = grads
        return grads

    @torch.no_grad()
    def graph_step(self):
        """The capturable step body: device step bump + ONE kernel.
        Must run inside the captured region, after backward."""
        ext = get_ext(required=True)
        g = self._graph
        g["step"].add_(1)
        group = self.param_groups[0]
        b1, b2 = group["betas"]
        table, nchunks, esize = g["table"]
        ext.fused_adam_graph(table, nchunks, esize, g["lr"], b1, b2,
                             group["eps"], group["weight_decay"],
                             g["step"])

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            ps, gs, ms, vs = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if len(st) == 0:
                    st["step"] = 0
                    st["m"] = torch.zeros_like(p, dtype=torch.float32)
                    st["v"] = torch.zeros_like(p, dtype=torch.float32)
                    if p.dtype == torch.bfloat16 and p.is_cuda:
                        st["master"] = p.detach().float().clone()
                st["step"] += 1
                ps.append(p)
                gs.append(p.grad)
                ms.append(st["m"])
                vs.append(st["v"])
            if not ps:
                continue
            b1, b2 = group["betas"]
            step = self.state[ps[0]]["step"]
            bc1 = 1 - b1 ** step
            bc2 = 1 - b2 ** step
            if _on_gpu(ps):
                ext = get_ext(required=True)
                masters = ([self.state[p]["master"] for p in ps]
                           if ps[0].dtype == torch.bfloat16 else [])
                ext.fused_adam(ps, gs, ms, vs, masters, group["lr"], b1, b2,
                               group["eps"], group["weight_decay"], bc1, bc2)
            else:
                for p, g, m, v in zip(ps, gs, ms, vs):
                    gf = g.float()
                    if group["weight_decay"]:
                        gf = gf.add(p.float(), alpha=group["weight_decay"])
                    m.mul_(b1).add_(gf, alpha=1 - b1)
                    v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
                    denom = (v / bc2).sqrt_().add_(group["eps"])
                    p.add_(((m / bc1) / denom).to(p.dtype),
                           alpha=-group["lr"])
        return loss


class FusedSGD(Optimizer):
    def __init__(self, params, lr=1e-2, momentum=0.0, weight_decay=0.0,
                 nesterov=False):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            ps, gs, bufs = [], [], []
            mom = group["momentum"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if mom and "momentum_buffer" not in st:
                    st["momentum_buffer"] = torch.zeros_like(
                        p, dtype=torch.float32)
                if p.dtype == torch.bfloat16 and p.is_cuda and \
                        "master" not in st:
                    st["master"] = p.detach().float().clone()
                ps.append(p)
                gs.append(p.grad)
                bufs.append(st["momentum_buffer"] if mom
                            else torch.zeros(0))
            if not ps:
                continue
            if _on_gpu(ps):
                ext = get_ext(required=True)
                masters = ([self.state[p]["master"] for p in ps]
                           if ps[0].dtype == torch.bfloat16 else [])
                ext.fused_sgd(ps, gs, bufs, masters, group["lr"], mom,
                              group["weight_decay"],
                              bool(group["nesterov"]))
            else:
                for p, g, b in zip(ps, gs, bufs):
                    gf = g.float()
                    if group["weight_decay"]:
                        gf = gf.add(p.float(), alpha=group["weight_decay"])
                    if mom:
                        b.mul_(mom).add_(gf)
                        gf = gf.add(b, alpha=mom) if group["nesterov"] else b
                    p.add_(gf.to(p.dtype), alpha=-group["lr"])
        return loss


class FusedLAMB(Optimizer):
    """LAMB (You et al.) with the per-tensor trust ratio
    clamp(|w| / |update|). Parity: torch_optimizer.Lamb used by the BERT
    example (examples/bert/provider.py:49-50)."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-6,
                 weight_decay=0.0, clamp_trust=10.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, clamp_trust=clamp_trust)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            b1, b2 = group["betas"]
            ps, gs, ms, vs = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if len(st) == 0:
                    st["step"] = 0
                    st["m"] = torch.zeros_like(p, dtype=torch.float32)
                    st["v"] = torch.zeros_like(p, dtype=torch.float32)
                    if p.dtype == torch.bfloat16 and p.is_cuda:
                        st["master"] = p.detach().float().clone()
                st["step"] += 1
                ps.append(p)
                gs.append(p.grad)
                ms.append(st["m"])
                vs.append(st["v"])
            if not ps:
                continue
            step = self.state[ps[0]]["step"]
            bc1 = 1 - b1 ** step
            bc2 = 1 - b2 ** step
            if _on_gpu(ps):
                ext = get_ext(required=True)
                masters = ([self.state[p]["master"] for p in ps]
                           if ps[0].dtype == torch.bfloat16 else [])
                ext.fused_lamb(ps, gs, ms, vs, masters, group["lr"], b1, b2,
                               group["eps"], group["weight_decay"], bc1, bc2,
                               group["clamp_trust"])
            else:
                for p, g, m, v in zip(ps, gs, ms, vs):
                    gf = g.float()
                    m.mul_(b1).add_(gf, alpha=1 - b1)
                    v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
                    upd = (m / bc1) / ((v / bc2).sqrt() + group["eps"])
                    if group["weight_decay"]:
                        upd = upd.add(p.float(), alpha=group["weight_decay"])
                    wn = p.float().norm()
                    un = upd.norm()
                    trust = torch.where(
                        (wn > 0) & (un > 0),
                        (wn / un).clamp(max=group["clamp_trust"]),
                        torch.ones_like(wn))
                    p.add_((trust * upd).to(p.dtype), alpha=-group["lr"])
        return loss
