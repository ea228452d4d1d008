"""minGPT sorter (parity: reference examples/sorter/provider.py): a
gpt-nano-scale decoder learns to sort 6-token sequences."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from examples.common import node_name  # noqa: E402

import torch  # noqa: E402
from torch.utils.data import DataLoader  # noqa: E402

from ravnest_amd import Node, Trainer, set_seed  # noqa: E402
from examples.sorter.dataset import SortDataset  # noqa: E402

set_seed(42)


def loss_fn(preds, targets):
    return torch.nn.functional.cross_entropy(
        preds.reshape(-1, preds.size(-1)).float(),
        targets[1].reshape(-1), ignore_index=-1)


if __name__ == "__main__":
    name, base_dir = node_name()
    g = torch.Generator().manual_seed(42)
    loader = DataLoader(SortDataset("train"), batch_size=64, shuffle=True,
                        generator=g)
    node = Node(name=name, base_dir=base_dir,
                optimizer=torch.optim.Adam,
                optimizer_params={"lr": 5e-4},
                criterion=loss_fn,
                labels=DataLoader(SortDataset("train"), batch_size=64,
                                  shuffle=True,
                                  generator=torch.Generator().manual_seed(42)),
                update_frequency=1)
    node.start()
    trainer = Trainer(node=node, train_loader=loader, epochs=1,
                      batch_size=64, save=True)
    trainer.train()

    # clean shutdown: root drains + cascades STOP; every rank closes its
    # channels (prevents the gloo teardown abort on live recv threads)
    if node.node_type.value == "root":
        node.stop_cluster()
    node.stop()
