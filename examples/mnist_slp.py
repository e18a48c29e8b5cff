"""MNIST SLP training with SynchronousSGDOptimizer (CPU plumbing config).

BASELINE.json config 1: "MNIST SLP SynchronousSGDOptimizer np=2 on CPU via
kungfu-run". Uses synthetic MNIST-shaped data (no network access for the
real dataset); convergence on a fixed synthetic task is still meaningful:
the model must fit a linear teacher.
"""
import argparse

import torch
import torch.nn.functional as F

import kungfu_amd as kf
from kungfu_amd.models import SLP
from kungfu_amd.ops import broadcast_model
from kungfu_amd.optimizers import SynchronousSGDOptimizer


def synthetic_mnist(n, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n, 1, 28, 28, generator=g)
    teacher = torch.randn(28 * 28, 10, generator=torch.Generator()
                          .manual_seed(1234))
    y = (x.flatten(1) @ teacher).argmax(1)
    return x, y


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n-epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--lr", type=float, default=0.2)
    p.add_argument("--restart", type=int, default=0)
    args = p.parse_args()

    kf.init(with_torch=False)
    torch.manual_seed(0)
    model = SLP()
    broadcast_model(model)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9))

    # shard data by rank (reference datasets/adaptor.py sharding)
    x, y = synthetic_mnist(4096, seed=42)
    shard = x.shape[0] // kf.size()
    lo = kf.rank() * shard
    x, y = x[lo:lo + shard], y[lo:lo + shard]

    for epoch in range(args.n_epochs):
        perm = torch.randperm(x.shape[0])
        total, correct, loss_sum = 0, 0, 0.0
        for i in range(0, x.shape[0], args.batch_size):
            idx = perm[i:i + args.batch_size]
            xb, yb = x[idx], y[idx]
            opt.zero_grad()
            out = model(xb)
            loss = F.cross_entropy(out, yb)
            loss.backward()
            opt.step()
            loss_sum += float(loss) * len(idx)
            correct += int((out.argmax(1) == yb).sum())
            total += len(idx)
        print("rank=%d epoch=%d loss=%.4f acc=%.3f" %
              (kf.rank(), epoch, loss_sum / total, correct / total),
              flush=True)
    acc = correct / total
    print("FINAL rank=%d acc=%.3f" % (kf.rank(), acc), flush=True)
    assert acc > 0.3, "did not learn"  # random = 0.1
    kf.finalize()


if __name__ == "__main__":
    main()
