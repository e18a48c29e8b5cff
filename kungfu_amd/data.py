"""Elastic dataset sharding.

Reference parity: srcs/python/kungfu/tensorflow/v1/datasets/adaptor.py —
shard the sample stream by (rank, size) with a mutable offset so the shard
assignment follows cluster resizes mid-epoch.
"""
import torch
from torch.utils.data import Sampler

import kungfu_amd as kf


class ElasticShardSampler(Sampler):
    """Shards indices across the current cluster; after a resize call
    set_progress() with the globally-synced sample offset and the sampler
    continues from there with the new (rank, size)."""

    def __init__(self, dataset_len, seed=0):
        self.n = int(dataset_len)
        self.seed = seed
        self.offset = 0  # samples already consumed cluster-wide this epoch
        self.epoch = 0

    def set_progress(self, offset, epoch=None):
        self.offset = int(offset) % self.n
        if epoch is not None:
            self.epoch = int(epoch)

    def __iter__(self):
        g = torch.Generator().manual_seed(self.seed + self.epoch)
        perm = torch.randperm(self.n, generator=g).tolist()
        rank, size = kf.rank(), kf.size()
        for i in range(self.offset + rank, self.n, size):
            yield perm[i]

    def __len__(self):
        size = kf.size()
        rank = kf.rank()
        remaining = max(0, self.n - self.offset)
        return (remaining - rank + size - 1) // size
