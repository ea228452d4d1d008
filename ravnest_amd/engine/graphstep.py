"""hipGraph capture of the fused training step (fwd + loss + bwd).

The fused single-stage path (root==leaf replica, the bench/DP
configuration) launches ~800 kernels per step; at ~90 ms steps the
launch+dispatch overhead is ~2.4 ms (rocprofv3 --stats, profiles/).
Capturing forward + loss + backward into one hipGraph replays the whole
chain from a single launch. MI355X-first design note: this replaces the
reference's tracing-compiler approach to launch overhead with HIP graphs
(the runtime facility the hardware stack provides).

Scope & correctness:
- Only the no-input-grad, update_frequency=1, versioning-off fused step
  is captured (exactly the bench path). Everything stateful stays
  OUTSIDE the graph: the optimizer step (its bias correction is computed
  host-side per step), DP parameter averaging, loss logging.
- Grad buffers: warmup ends with grads set to None, so capture records
  plain assignments into graph-pool tensors; every replay overwrites the
  same buffers and `param.grad` keeps pointing at them. The engine must
  NOT zero grads between graphed steps (assignment semantics make
  zeroing unnecessary and `set_to_none` would orphan the references).
- RNG: dropout draws its philox seed from the CPU generator (frozen at
  capture); the kernel XORs in a device counter (ops/rng.py) that the
  captured region increments once per replay, so masks stay fresh and
  same-seed runs stay reproducible.
- Any capture failure permanently falls back to the eager path.
"""
from __future__ import annotations

import torch

from ..ops import rng


def _sig(tensors):
    return tuple((tuple(t.shape), t.dtype) for t in tensors)


class GraphedTrainStep:
    """Per-shape-signature hipGraph cache for the fused train step."""

    WARMUP = 3

    def __init__(self, model, criterion, device, optimizer=None):
        self.model = model
        self.criterion = criterion
        self.device = device
        # optimizer captured INTO the graph when it supports device-side
        # lr/step buffers (FusedAdam.graph_step). OFF by default: the
        # capture needs pre-existing grad buffers, which flips autograd
        # from assign-into-pool to accumulate (temp + add per weight
        # grad) — measured 1457 vs 1561 samples/s on BERT-base, so the
        # eager one-kernel optimizer wins. RAVNEST_GRAPH_OPT=1 enables
        # full-step capture (single launch incl. update) for launch-
        # latency-dominated regimes (small models / many tiny steps).
        import os
        use_opt = os.environ.get("RAVNEST_GRAPH_OPT", "0") == "1"
        self.optimizer = (optimizer if use_opt and
                          hasattr(optimizer, "graph_step") else None)
        self.opt_captured = False
        self.graphs = {}
        self.pool = None
        self.failed = False

    def run(self, args, targets):
        """Replay (or build) the captured step for this shape signature.

        Returns the static loss tensor (device) or None when the step
        cannot be graphed — caller must then run the eager path.
        """
        if self.failed or self.device.type != "cuda":
            return None
        # supported: all-tensor args and a single tensor target (the
        # fused bench configuration); anything else stays eager
        if not all(torch.is_tensor(a) for a in args) or \
                not torch.is_tensor(targets):
            return None
        arg_ts = list(args)
        tgt_ts = [targets]
        key = (_sig(arg_ts), _sig(tgt_ts))
        entry = self.graphs.get(key)
        if entry is None:
            try:
                entry = self._capture(arg_ts, tgt_ts)
            except Exception:
                self.failed = True
                # leave grads in a clean eager state
                self.model.zero_grad(set_to_none=True)
                return None
            self.graphs[key] = entry
        graph, s_args, s_tgts, s_loss, grad_refs = entry
        if self.opt_captured:
            self.optimizer.sync_lr()  # outside the graph: live schedules
        for dst, src in zip(s_args, arg_ts):
            dst.copy_(src, non_blocking=True)
        for dst, src in zip(s_tgts, tgt_ts):
            dst.copy_(src, non_blocking=True)
        graph.replay()
        # re-attach the capture-time grad buffers: an interleaved eager
        # step may have zero_grad(set_to_none=True)-detached them; the
        # replay fully overwrote the buffers either way
        for param, gref in grad_refs:
            param.grad = gref
        return s_loss

    # ------------------------------------------------------------------
    def _capture(self, arg_ts, tgt_ts):
        s_args = [t.clone() for t in arg_ts]
        s_tgts = [t.clone() for t in tgt_ts]
        self.model.train()
        rng.device_seed_counter(self.device)  # materialize pre-capture

        if self.optimizer is not None and not self.opt_captured:
            self.opt_captured = self.optimizer.enable_graph_capture(
                self.device)

        def step(with_opt=False, stable_grads=None):
            rng.bump_seed_counter(self.device)
            if with_opt:
                # stable (non-pool) grad buffers: zero, accumulate, apply
                torch._foreach_zero_(stable_grads)
            out = self.model(*s_args)
            loss = self.criterion(out, s_tgts[0])
            loss.backward()
            if with_opt:
                self.optimizer.graph_step()
            return loss

        # warmup on a side stream (per the torch.cuda.graph recipe):
        # autotuners/allocators settle before capture
        side = torch.cuda.Stream(self.device)
        side.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(side):
            for _ in range(self.WARMUP):
                step()
        torch.cuda.current_stream(self.device).wait_stream(side)
        # opt-captured mode: stable pre-allocated grad buffers (zeroed
        # and accumulated INSIDE the graph); else grads stay None so
        # capture records plain assignments into pool buffers
        self.model.zero_grad(set_to_none=True)
        stable_grads = None
        if self.opt_captured:
            # built ONCE: a second shape-signature capture must reuse the
            # same stable buffers (earlier graphs recorded their pointers)
            stable_grads = self.optimizer._graph.get("grads")
            if stable_grads is None:
                stable_grads = self.optimizer.build_graph_table()

        graph = torch.cuda.CUDAGraph()
        if self.pool is None:
            with torch.cuda.graph(graph):
                s_loss = step(self.opt_captured, stable_grads)
            self.pool = graph.pool()
        else:
            with torch.cuda.graph(graph, pool=self.pool):
                s_loss = step(self.opt_captured, stable_grads)
        grad_refs = [(p, p.grad) for p in self.model.parameters()
                     if p.grad is not None]
        return graph, s_args, s_tgts, s_loss, grad_refs
