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


def mnist_idx(data_dir, train=True):
    """Load real MNIST from local IDX files (reference
    kungfu/tensorflow/v1/helpers/mnist.py reads the same format after
    download; this environment has no network, so files must already be
    on disk — use synthetic_mnist otherwise). Accepts gzipped or raw
    idx files with the standard names."""
    import gzip
    import os
    import struct

    import numpy as np

    def read_idx(name):
        for n in (name, name + ".gz"):
            p = os.path.join(data_dir, n)
            if os.path.exists(p):
                op = gzip.open if n.endswith(".gz") else open
                with op(p, "rb") as f:
                    data = f.read()
                magic, = struct.unpack(">I", data[:4])
                ndim = magic & 0xff
                dims = struct.unpack(">" + "I" * ndim,
                                     data[4:4 + 4 * ndim])
                arr = np.frombuffer(data, dtype=np.uint8,
                                    offset=4 + 4 * ndim)
                return arr.reshape(dims)
        raise FileNotFoundError("%s(.gz) not in %s" % (name, data_dir))

    prefix = "train" if train else "t10k"
    x = read_idx("%s-images-idx3-ubyte" % prefix)
    y = read_idx("%s-labels-idx1-ubyte" % prefix)
    xt = torch.from_numpy(x.copy()).float().div_(255.0).unsqueeze(1)
    yt = torch.from_numpy(y.copy()).long()
    return torch.utils.data.TensorDataset(xt, yt)
