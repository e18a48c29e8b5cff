"""Single-layer perceptron for MNIST-shaped plumbing tests.

Reference parity: the `slp-mnist` fake model / test_mnist_slp.py fixtures
(tests/go/fakemodel/fakemodel.go:12-17).
"""
import torch.nn as nn


class SLP(nn.Module):
    def __init__(self, in_features=28 * 28, classes=10):
        super().__init__()
        self.fc = nn.Linear(in_features, classes)

    def forward(self, x):
        return self.fc(x.flatten(1))
