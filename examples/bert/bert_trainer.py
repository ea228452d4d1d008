"""Custom Trainer feeding kwargs microbatches (parity: reference
examples/bert/bert_trainer.py — the documented custom-Trainer extension
point, docs/features.rst "Custom Trainers")."""
from ravnest_amd import Trainer


class BertTrainer(Trainer):
    def train(self):
        self.prelim_checks()
        if self.passive:
            return
        for epoch in range(self.epochs):
            for batch in self.train_loader:
                self.node.forward_compute(
                    input_ids=batch["input_ids"],
                    attention_mask=batch["attention_mask"])
                self.n_forwards += 1
            self.node.wait_for_backwards()
            if self.lr_scheduler is not None:
                self.lr_scheduler.step()
        self.node.comm_session.parallel_ring_reduce()
        if self.save:
            self.node.trigger_save_submodel()
