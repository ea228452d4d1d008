"""CNN on sklearn-digits, 3-stage pipeline — the canonical walkthrough
workload (reference examples/cnn/provider.py; real data, ships with
sklearn). Run cluster_formation.py first, then one provider per terminal:
    python examples/cnn/provider.py --name node_0   # root
    python examples/cnn/provider.py --name node_1   # stem
    python examples/cnn/provider.py --name node_2   # leaf
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from examples.common import node_name  # noqa: E402

import numpy as np  # noqa: E402
import torch  # noqa: E402
from torch.utils.data import DataLoader  # noqa: E402
from sklearn import datasets  # noqa: E402
from sklearn.model_selection import train_test_split  # noqa: E402

from ravnest_amd import Node, Trainer, set_seed  # noqa: E402

set_seed(42)


def digits_loaders(batch_size: int = 64, holdout: float = 0.4,
                   split_seed: int = 1, order_seed: int = 42):
    """sklearn-digits as (train, val) DataLoaders of (image, one_hot)
    pairs. Same workload *config* as the reference walkthrough (8x8
    images, MSE vs one-hot, 60/40 split, seeded shuffle order so every
    pipeline rank that wraps these loaders sees identical batches).
    Torch-native prep: one_hot via torch, no numpy staging."""
    raw = datasets.load_digits()
    images = torch.as_tensor(raw.images, dtype=torch.float32).unsqueeze(1)
    onehot = torch.nn.functional.one_hot(
        torch.as_tensor(raw.target, dtype=torch.long), 10).float()
    idx_train, idx_val = train_test_split(
        np.arange(len(images)), test_size=holdout, random_state=split_seed)
    g = torch.Generator()
    g.manual_seed(order_seed)
    train = DataLoader(
        [(images[i], onehot[i]) for i in idx_train],
        generator=g, shuffle=True, batch_size=batch_size)
    val = DataLoader(
        [(images[i], onehot[i]) for i in idx_val],
        shuffle=False, batch_size=batch_size)
    return train, val


def loss_fn(preds, targets):
    # leaf criterion: targets is the (X, one_hot) pair from the label feed
    return torch.nn.functional.mse_loss(preds, targets[1])


if __name__ == "__main__":
    name, base_dir = node_name()
    train_loader, val_loader = digits_loaders()
    node = Node(name=name, base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=loss_fn,
                labels=train_loader,
                test_labels=val_loader)
    node.start()
    trainer = Trainer(node=node, train_loader=train_loader,
                      val_loader=val_loader, val_freq=64, epochs=100,
                      batch_size=64, inputs_dtype=torch.float32)
    trainer.train()
    trainer.evaluate()

    # clean shutdown: root drains + cascades STOP; every rank closes its
    # channels (prevents the gloo teardown abort on live recv threads)
    if node.node_type.value == "root":
        node.stop_cluster()
    node.stop()
