"""BERT-base MLM pretraining, 4-stage pipeline, LAMB + grad accumulation
(parity: reference examples/bert/provider.py — WikiText replaced by
synthetic token batches: no network in this environment)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from examples.common import node_name  # noqa: E402

import torch  # noqa: E402

from ravnest_amd import Node, set_seed  # noqa: E402
from ravnest_amd.models import BertConfig, BertForMLM  # noqa: E402
from ravnest_amd.ops import CrossEntropyLoss, FusedLAMB  # noqa: E402
from examples.bert.bert_trainer import BertTrainer  # noqa: E402

SEQ, BATCH, NBATCH = 512, 8, 64

set_seed(42)


def synthetic_batches(vocab):
    g = torch.Generator().manual_seed(42)
    out = []
    for _ in range(NBATCH):
        ids = torch.randint(0, vocab, (BATCH, SEQ), generator=g)
        labels = ids.clone()
        labels[torch.rand(ids.shape, generator=g) > 0.15] = -100
        out.append({"input_ids": ids,
                    "attention_mask": torch.ones_like(ids),
                    "labels": labels})
    return out


if __name__ == "__main__":
    name, base_dir = node_name()
    cfg = BertConfig.base()
    batches = synthetic_batches(cfg.vocab_size)
    crit = CrossEntropyLoss(-100)
    node = Node(name=name, base_dir=base_dir,
                optimizer=FusedLAMB,
                optimizer_params={"lr": 1.76e-3},
                criterion=lambda preds, b: crit(preds, b["labels"]),
                labels=batches,
                update_frequency=16)
    node.start()
    trainer = BertTrainer(node=node, train_loader=batches, epochs=45,
                          batch_size=BATCH)
    trainer.train()

    # clean shutdown: root drains + cascades STOP; every rank closes its
    # channels (prevents the gloo teardown abort on live recv threads)
    if node.node_type.value == "root":
        node.stop_cluster()
    node.stop()
