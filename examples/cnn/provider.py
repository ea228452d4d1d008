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


def to_categorical(x, n_col=None):
    n_col = n_col or (np.amax(x) + 1)
    one_hot = np.zeros((x.shape[0], n_col))
    one_hot[np.arange(x.shape[0]), x] = 1
    return one_hot


def preprocess_dataset():
    data = datasets.load_digits()
    X, y = data.data, to_categorical(data.target.astype("int"))
    X_train, X_test, y_train, y_test = train_test_split(
        X, y, test_size=0.4, random_state=1)
    X_train = X_train.reshape((-1, 1, 8, 8)).astype("float32")
    X_test = X_test.reshape((-1, 1, 8, 8)).astype("float32")
    g = torch.Generator()
    g.manual_seed(42)
    train = DataLoader(list(zip(torch.tensor(X_train),
                                torch.tensor(y_train, dtype=torch.float32))),
                       generator=g, shuffle=True, batch_size=64)
    val = DataLoader(list(zip(torch.tensor(X_test),
                              torch.tensor(y_test, dtype=torch.float32))),
                     shuffle=False, batch_size=64)
    return train, val


def loss_fn(preds, targets):
    return torch.nn.functional.mse_loss(preds, targets[1])


if __name__ == "__main__":
    name, base_dir = node_name()
    train_loader, val_loader = preprocess_dataset()
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
