"""Synthetic dataset helpers.

Reference parity: kungfu/tensorflow/v1/helpers/{mnist,cifar,imagenet}.py —
but this environment has no network, so these generate dataset-shaped
synthetic tensors (deterministic by seed) and shard them elastically.
"""
import torch

from kungfu_amd.data import ElasticShardSampler


def synthetic_mnist(n=4096, seed=42):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n, 1, 28, 28, generator=g)
    teacher = torch.randn(28 * 28, 10,
                          generator=torch.Generator().manual_seed(1234))
    y = (x.flatten(1) @ teacher).argmax(1)
    return torch.utils.data.TensorDataset(x, y)


def synthetic_cifar10(n=2048, seed=43):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (n,), generator=g)
    return torch.utils.data.TensorDataset(x, y)


def synthetic_imagenet(n=512, seed=44, size=224):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 3, size, size, generator=g)
    y = torch.randint(0, 1000, (n,), generator=g)
    return torch.utils.data.TensorDataset(x, y)


def elastic_loader(dataset, batch_size, seed=0, **kw):
    """DataLoader sharded by the current cluster; reshards after resizes
    via sampler.set_progress()."""
    sampler = ElasticShardSampler(len(dataset), seed=seed)
    loader = torch.utils.data.DataLoader(dataset, batch_size=batch_size,
                                         sampler=sampler, **kw)
    return loader, sampler
