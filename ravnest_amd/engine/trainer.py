"""User-facing training loop driver.

API parity with the reference Trainer (ravnest/trainer.py:6-126): active
only on the root (stem/leaf ranks park in `prelim_checks`), epoch loop
feeding `node.forward_compute`, validation via `no_grad_forward_compute`,
final DP parameter averaging and optional submodel save. Subclassable for
custom loops (the reference's BERT example subclasses it to feed kwargs).
"""
from __future__ import annotations

import time

import numpy as np
import torch

from ..strings import NodeTypes


class Trainer:
    def __init__(self, node=None, lr_scheduler=None, lr_scheduler_params=None,
                 train_loader=None, val_loader=None, val_freq=1, save=False,
                 epochs=1, batch_size=64, step_size=1, inputs_dtype=None):
        self.node = node
        self.passive = node.node_type in (NodeTypes.STEM, NodeTypes.LEAF) \
            and not node.fused
        if self.passive:
            return
        self.train_loader = train_loader
        self.val_loader = val_loader
        self.val_freq = val_freq
        self.save = save
        self.epochs = epochs
        self.batch_size = batch_size
        self.step_size = step_size
        self.n_forwards = 0
        self.inputs_dtype = inputs_dtype
        self.lr_scheduler = None
        if lr_scheduler is not None:
            self.lr_scheduler = lr_scheduler(self.node.optimizer,
                                             **(lr_scheduler_params or {}))

    def prelim_checks(self):
        """Stem/leaf ranks have no data feed: they serve the pipeline until
        stopped (parity trainer.py:54-57, but event-driven — the dispatch
        thread does the work; this thread just parks)."""
        if self.passive:
            while not self.node._stop.is_set():
                time.sleep(0.05)

    def _coerce(self, X):
        if not torch.is_tensor(X):
            X = torch.tensor(np.asarray(X))
        if self.inputs_dtype is not None and X.dtype != self.inputs_dtype:
            X = X.to(self.inputs_dtype)
        return X

    def train(self):
        self.prelim_checks()
        if self.passive:
            return
        t1 = time.time()
        self.n_forwards = 0
        for epoch in range(self.epochs):
            for batch in self.train_loader:
                X = batch[0] if isinstance(batch, (tuple, list)) else batch
                self.node.forward_compute(tensors=self._coerce(X))
                self.n_forwards += 1
                # mid-epoch validation cadence: every val_freq training
                # batches (reference trainer.py:21-22 documents this;
                # val_freq=1 keeps the once-per-epoch behavior below)
                if self.val_loader is not None and self.val_freq and \
                        self.val_freq > 1 and \
                        self.n_forwards % self.val_freq == 0:
                    self._run_validation()
            if self.val_loader is not None and (not self.val_freq
                                                or self.val_freq <= 1):
                self._run_validation()
            self.node.wait_for_backwards()
            if self.lr_scheduler is not None:
                self.lr_scheduler.step()
        self.node.wait_for_backwards()
        self.node.comm_session.parallel_ring_reduce()
        self.train_time = time.time() - t1
        if self.save:
            self.node.trigger_save_submodel()

    def _run_validation(self):
        self.node.wait_for_backwards()
        for batch in self.val_loader:
            X = batch[0] if isinstance(batch, (tuple, list)) else batch
            self.node.no_grad_forward_compute(
                tensors=self._coerce(X), output_type="val_accuracy")

    def pred(self, data):
        if self.passive:
            return
        return self.node.no_grad_forward_compute(
            tensors=self._coerce(data), output_type="accuracy")

    def evaluate(self):
        if self.passive:
            return
        self.node.wait_for_backwards()
        for batch in self.val_loader:
            X = batch[0] if isinstance(batch, (tuple, list)) else batch
            self.node.no_grad_forward_compute(
                tensors=self._coerce(X), output_type="val_accuracy")
