"""Versioned asynchronous compute engine: no-grad forward with RNG/input
capture, parameter-version snapshots, recompute-then-backward, speculative
recompute.

This is the signature mechanism of the reference (ravnest/compute.py:10-271,
SURVEY.md section 2.4), re-built device-native for MI355X:

* forwards run under ``torch.no_grad`` and capture inputs + CPU/HIP RNG
  state per forward-pass id (fpid) — parity compute.py:53-122;
* optimizer steps bump ``current_version`` and snapshot a device-resident
  parameter clone (288 GB HBM3E holds a deep ring of these) — parity
  compute.py:47-51; unlike the reference (which clones every backward,
  compute.py:190-199) snapshots are taken only when parameters actually
  change, and are refcounted by outstanding fpids;
* backward first RECOMPUTES the forward with the exact historical
  parameter version and RNG (``param.data`` swap keeps optimizer references
  valid — parity utils.py:119-125 load_state_dict_conserve_versions,
  compute.py:214-271), restores current parameters, then runs
  ``torch.autograd.backward`` over the recomputed outputs;
* after each backward the next fpid's recompute is kicked off on a
  background thread, joined before the next compute op — parity
  compute.py:204-206,54-56,95-97,136-138.
"""
from __future__ import annotations

import os
import threading
from dataclasses import dataclass

import torch

from ..utils import current_rng_states, restore_rng_states, trace_range


@dataclass
class _FpidRecord:
    version: int
    rng: dict
    args: list  # positional args as captured (tensors or None)
    needs_grad: list  # per-arg bool: will we return a grad for it
    recomputed_args: list | None = None
    recomputed_outputs: tuple | None = None


class ComputeEngine:
    def __init__(self, model: torch.nn.Module,
                 optimizer: torch.optim.Optimizer | None,
                 device: torch.device,
                 update_frequency: int = 1,
                 criterion=None,
                 loss_filename: str = "losses.txt",
                 amp_dtype: torch.dtype | None = None,
                 versioning: bool = True):
        self.model = model
        self.optimizer = optimizer
        self.device = device
        self.update_frequency = max(1, update_frequency)
        self.criterion = criterion
        self.loss_filename = loss_filename
        self.amp_dtype = amp_dtype
        # versioning=False (single-fused-stage training) skips the
        # per-step parameter snapshot clones: no recompute ever happens
        self.versioning = versioning
        # hipGraph capture of the fused step (fwd+loss+bwd single
        # launch; see graphstep.py). Only safe when no recompute/grad
        # accumulation semantics are in play.
        self._graph_step = None
        if (not versioning and device.type == "cuda"
                and self.update_frequency == 1 and criterion is not None
                and os.environ.get("RAVNEST_CUDA_GRAPH", "1") == "1"):
            from .graphstep import GraphedTrainStep
            self._graph_step = GraphedTrainStep(model, criterion, device,
                                                optimizer=optimizer)

        # invoked right before every optimizer step / graph replay that
        # steps: the node uses it to JOIN an overlapped DP averaging
        # (comm/collectives.py AsyncReducer) so the averaged values land
        # before new updates are applied
        self.pre_step_hook = None

        self.current_version = 0
        self.version_to_param: dict[int, list[torch.Tensor]] = {}
        self.version_refs: dict[int, int] = {}
        self.fpids: dict[int, _FpidRecord] = {}
        self.n_backwards = 0
        self.file_loss = 0.0
        self._recompute_thread: threading.Thread | None = None
        self._lock = threading.Lock()

        self._params = list(self.model.parameters())
        self._snapshot_current()

    def _autocast(self):
        if self.amp_dtype is not None and self.device.type == "cuda":
            return torch.autocast("cuda", dtype=self.amp_dtype)
        import contextlib
        return contextlib.nullcontext()

    # ------------------------------------------------------------------
    # version bookkeeping
    # ------------------------------------------------------------------
    def _snapshot_current(self):
        if not self.versioning:
            return
        snap = [torch.empty_like(p) for p in self._params]
        self._fast_copy([p.detach() for p in self._params], snap)
        self.version_to_param[self.current_version] = snap
        self.version_refs.setdefault(self.current_version, 0)

    def _fast_copy(self, srcs, dsts):
        """Multi-tensor device copy: one kernel launch for the whole
        parameter set (vs ~#params hipMemcpy calls)."""
        if srcs and srcs[0].is_cuda:
            from ..ops import get_ext
            ext = get_ext(required=False)
            if ext is not None:
                ext.fused_copy(srcs, dsts)
                return
        for s, d in zip(srcs, dsts):
            d.copy_(s)

    def latest_state_snapshot(self) -> dict:
        """Latest parameter snapshot for peers to pull (parity:
        latest_weights_buffer, reference compute.py:47-51)."""
        with self._lock:
            if not self.versioning:
                return {"version": self.current_version,
                        "params": [p.detach().clone() for p in self._params]}
            return {
                "version": self.current_version,
                "params": [t.clone() for t in
                           self.version_to_param[self.current_version]],
            }

    def load_param_list(self, params: list[torch.Tensor]):
        """Install pulled weights (elastic join / latest-weights pull)."""
        with torch.no_grad():
            for p, src in zip(self._params, params):
                p.data.copy_(src.to(p.device))
        self.bump_version()

    def _release_version(self, v: int):
        self.version_refs[v] -= 1
        if self.version_refs[v] == 0 and v != self.current_version:
            del self.version_refs[v]
            del self.version_to_param[v]

    def bump_version(self):
        """Called whenever parameters change (optimizer step, DP averaging,
        weight pull)."""
        old = self.current_version
        self.current_version += 1
        if not self.versioning:
            return
        self._snapshot_current()
        if self.version_refs.get(old, 0) == 0:
            self.version_refs.pop(old, None)
            self.version_to_param.pop(old, None)

    def _swap_params_to(self, version: int):
        snap = self.version_to_param[version]
        for p, s in zip(self._params, snap):
            p.data = s

    def _restore_params(self):
        # restore live (current) parameter storages: current snapshot is a
        # clone, but live training must continue on the ORIGINAL storages
        # so optimizer state stays bound. Keep originals separately.
        for p, orig in zip(self._params, self._orig_data):
            p.data = orig

    # ------------------------------------------------------------------
    # forward (training, no_grad + capture)
    # ------------------------------------------------------------------
    def forward(self, fpid: int, args: list, needs_grad: list) -> tuple:
        """No-grad forward with capture. `args` are the stage's positional
        inputs (device tensors or python constants); `needs_grad[i]` marks
        inputs whose grad must be produced at backward time."""
        self.join_recompute()
        if not self.model.training:
            self.model.train()
        rng = current_rng_states(self.device)
        rec = _FpidRecord(version=self.current_version, rng=rng,
                          args=[a.detach().clone() if torch.is_tensor(a) else a
                                for a in args],
                          needs_grad=list(needs_grad))
        with torch.no_grad(), self._autocast(), trace_range(f"fwd:{fpid}"):
            out = self.model(*args)
        outputs = out if isinstance(out, tuple) else (out,)
        with self._lock:
            self.fpids[fpid] = rec
            self.version_refs[rec.version] = self.version_refs.get(rec.version, 0) + 1
        return outputs

    # ------------------------------------------------------------------
    # recompute + backward (root/stem stages)
    # ------------------------------------------------------------------
    def _recompute(self, fpid: int):
        rec = self.fpids[fpid]
        if rec.recomputed_outputs is not None:
            return
        args = []
        for a, ng in zip(rec.args, rec.needs_grad):
            if torch.is_tensor(a) and a.is_floating_point():
                a = a.detach().clone().requires_grad_(True) if ng else a
            args.append(a)
        self._orig_data = [p.data for p in self._params]
        self._swap_params_to(rec.version)
        try:
            if not self.model.training:
                self.model.train()
            devices = [self.device] if self.device.type == "cuda" else []
            with torch.random.fork_rng(devices=devices):
                restore_rng_states(rec.rng, self.device)
                with self._autocast():
                    out = self.model(*args)
        finally:
            self._restore_params()
        rec.recomputed_outputs = out if isinstance(out, tuple) else (out,)
        rec.recomputed_args = args

    def join_recompute(self):
        t = self._recompute_thread
        if t is not None and t.is_alive():
            t.join()

    def backward(self, fpid: int, grads_by_out_idx: dict[int, torch.Tensor],
                 speculative_next: bool = True
                 ) -> tuple[list[torch.Tensor | None], bool]:
        """Recompute fpid's forward at its historical version, backprop the
        received output grads, accumulate into .grad, and optimizer-step on
        the accumulation boundary.

        Returns (per-arg input grads aligned with the captured args — None
        where needs_grad was False, stepped_flag)."""
        self.join_recompute()
        rec = self.fpids[fpid]
        self._recompute(fpid)
        outs, grads = [], []
        for k, g in grads_by_out_idx.items():
            out = rec.recomputed_outputs[k]
            if out.grad_fn is None:
                continue
            outs.append(out)
            grads.append(g.to(out.dtype).to(out.device))
        if outs:
            with trace_range(f"bwd:{fpid}"):
                torch.autograd.backward(outs, grads)

        input_grads: list[torch.Tensor | None] = []
        for a, ng in zip(rec.recomputed_args, rec.needs_grad):
            if ng and torch.is_tensor(a):
                input_grads.append(a.grad if a.grad is not None
                                   else torch.zeros_like(a))
            else:
                input_grads.append(None)

        stepped = self._finish_backward(fpid, rec)
        if speculative_next and (fpid + 1) in self.fpids:
            self._recompute_thread = threading.Thread(
                target=self._recompute, args=(fpid + 1,), daemon=True)
            self._recompute_thread.start()
        return input_grads, stepped

    def _finish_backward(self, fpid: int, rec: _FpidRecord) -> bool:
        with self._lock:
            del self.fpids[fpid]
            self._release_version(rec.version)
        self.n_backwards += 1
        stepped = False
        if self.optimizer is not None and \
                self.n_backwards % self.update_frequency == 0:
            if self.pre_step_hook is not None:
                self.pre_step_hook()
            self.optimizer.step()
            self.optimizer.zero_grad(set_to_none=True)
            self.model.zero_grad(set_to_none=True)
            self.bump_version()
            stepped = True
        return stepped

    # ------------------------------------------------------------------
    # leaf: true forward + loss + backward at current weights
    # (parity: reference leaf_find_loss, compute.py:273-327)
    # ------------------------------------------------------------------
    def find_loss(self, fpid: int, args: list, needs_grad: list, targets
                  ) -> tuple[list[torch.Tensor | None], bool, float]:
        self.join_recompute()
        if not self.model.training:
            self.model.train()
        if self._graph_step is not None and not any(needs_grad) and \
                self.amp_dtype is None:
            # join a pending overlapped DP reduce BEFORE the replay: the
            # captured region may contain the optimizer step
            if self.pre_step_hook is not None:
                self.pre_step_hook()
            loss_t = self._graph_step.run(args, targets)
            if loss_t is not None:
                return self._finish_graphed_step(args, loss_t)
        if self._graph_step is not None and self._graph_step.graphs:
            # eager step after a capture: drop any stale graph-pool grad
            # buffers so this backward starts clean (the next replay
            # re-attaches its references)
            self.model.zero_grad(set_to_none=True)
        live_args = []
        for a, ng in zip(args, needs_grad):
            if ng and torch.is_tensor(a) and a.is_floating_point():
                a = a.detach().clone().requires_grad_(True)
            live_args.append(a)
        with self._autocast():
            out = self.model(*live_args)
            loss = self.criterion(out, targets)
        loss.backward()
        loss_val = float(loss.detach())
        self.file_loss += loss_val

        input_grads: list[torch.Tensor | None] = []
        for a, ng in zip(live_args, needs_grad):
            if ng and torch.is_tensor(a):
                input_grads.append(a.grad if a.grad is not None
                                   else torch.zeros_like(a))
            else:
                input_grads.append(None)

        self.n_backwards += 1
        stepped = False
        if self.optimizer is not None and \
                self.n_backwards % self.update_frequency == 0:
            if self.pre_step_hook is not None:
                self.pre_step_hook()
            self.optimizer.step()
            self.optimizer.zero_grad(set_to_none=True)
            self.model.zero_grad(set_to_none=True)
            self.bump_version()
            stepped = True
            if self.loss_filename:
                with open(self.loss_filename, "a") as f:
                    f.write(f"{round(self.file_loss, 4)}\n")
            self.file_loss = 0.0
        return input_grads, stepped, loss_val

    def _finish_graphed_step(self, args, loss_t):
        """Post-replay bookkeeping: optimizer step stays EAGER (Adam bias
        correction is computed host-side per step and must not freeze at
        capture time) and grads are NOT zeroed (the captured backward
        records assignments into stable graph-pool buffers that
        param.grad keeps referencing — see graphstep.py)."""
        # ENQUEUE the optimizer before reading the loss: float(loss_t)
        # synchronizes on the replay, and the optimizer's host-side chunk
        # building would otherwise run on an idle GPU. When the optimizer
        # was captured INTO the graph the replay already applied it; only
        # the host-side step mirror needs advancing (checkpoints).
        self.n_backwards += 1
        stepped = False
        if self._graph_step.opt_captured:
            self.optimizer.bump_host_steps()
            stepped = True
        elif self.optimizer is not None:
            self.optimizer.step()
            stepped = True
        loss_val = float(loss_t)
        self.file_loss += loss_val
        if stepped:
            if self.loss_filename:
                with open(self.loss_filename, "a") as f:
                    f.write(f"{round(self.file_loss, 4)}\n")
            self.file_loss = 0.0
        return [None] * len(args), stepped, loss_val

    # ------------------------------------------------------------------
    # eval forwards
    # ------------------------------------------------------------------
    def no_grad_forward(self, args: list) -> tuple:
        self.join_recompute()
        self.model.eval()
        with torch.no_grad(), self._autocast():
            out = self.model(*args)
        return out if isinstance(out, tuple) else (out,)
