"""MNIST MLP — the reference's demo model (SURVEY §2a "model fn"):
784 -> hidden -> 10, softmax-cross-entropy loss. Used by the CPU/gloo
plumbing config (BASELINE.json configs[0]) and the loss-parity gate."""

from __future__ import annotations

import torch
import torch.nn as nn

__all__ = ["MnistMLP", "mnist_mlp", "synthetic_mnist_batch"]


class MnistMLP(nn.Module):
    def __init__(self, hidden: int = 128, num_classes: int = 10,
                 in_features: int = 784):
        super().__init__()
        self.fc1 = nn.Linear(in_features, hidden)
        self.fc2 = nn.Linear(hidden, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.reshape(x.shape[0], -1)
        return self.fc2(torch.relu(self.fc1(x)))

    def loss(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        from rocps.ops.softmax_xent import softmax_cross_entropy_with_logits
        return softmax_cross_entropy_with_logits(self(x), y).mean()


def mnist_mlp(hidden: int = 128, **kw) -> MnistMLP:
    return MnistMLP(hidden=hidden, **kw)


def synthetic_mnist_batch(batch_size: int, device="cpu",
                          generator: torch.Generator = None):
    x = torch.rand(batch_size, 784, device=device, generator=generator)
    y = torch.randint(0, 10, (batch_size,), device=device, generator=generator)
    return x, y
