"""ResNet-50, 2-stage pipeline, SGD+momentum (parity: reference
examples/resnet50/provider.py — TinyImageNet replaced by synthetic
200-class 64x64 batches: no network in this environment)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from examples.common import node_name  # noqa: E402

import torch  # noqa: E402

from ravnest_amd import Node, Trainer, set_seed  # noqa: E402
from ravnest_amd.ops import FusedSGD  # noqa: E402

set_seed(42)
BATCH, NBATCH = 100, 50


def synthetic_loader():
    g = torch.Generator().manual_seed(42)
    data = []
    for _ in range(NBATCH):
        X = torch.randn(BATCH, 3, 64, 64, generator=g)
        y = torch.randint(0, 200, (BATCH,), generator=g)
        data.append((X, y))
    return data


def loss_fn(preds, targets):
    return torch.nn.functional.cross_entropy(preds.float(), targets[1])


if __name__ == "__main__":
    name, base_dir = node_name()
    loader = synthetic_loader()
    node = Node(name=name, base_dir=base_dir,
                optimizer=FusedSGD,
                optimizer_params={"lr": 0.01, "momentum": 0.9,
                                  "weight_decay": 5e-4},
                criterion=loss_fn,
                labels=loader)
    node.start()
    trainer = Trainer(node=node, train_loader=loader, epochs=5,
                      batch_size=BATCH, save=True)
    trainer.train()

    # clean shutdown: root drains + cascades STOP; every rank closes its
    # channels (prevents the gloo teardown abort on live recv threads)
    if node.node_type.value == "root":
        node.stop_cluster()
    node.stop()
