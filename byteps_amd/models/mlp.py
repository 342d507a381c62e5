"""MNIST MLP — the minimal end-to-end plumbing model (BASELINE.json
config 1: "MNIST MLP push_pull on CPU/gloo")."""

import torch.nn as nn


def mnist_mlp(hidden: int = 256, num_classes: int = 10) -> nn.Module:
    return nn.Sequential(
        nn.Flatten(),
        nn.Linear(28 * 28, hidden), nn.ReLU(),
        nn.Linear(hidden, hidden), nn.ReLU(),
        nn.Linear(hidden, num_classes))
