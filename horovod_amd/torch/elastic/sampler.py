"""ElasticSampler: shards a dataset across the current world and records
processed indices so a resumed epoch skips work already done.

Reference: horovod/torch/elastic/sampler.py:24-121.
"""
import math
import random

import torch.utils.data.distributed

from horovod_amd.torch.mpi_ops import is_initialized, rank, size


def _world():
    if is_initialized():
        return rank(), size()
    return 0, 1


class ElasticSampler(torch.utils.data.Sampler):
    def __init__(self, dataset, shuffle=True, seed=0):
        self.dataset = dataset
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = 0
        self.processed_indices = set()

        self.num_replicas = 0
        self.rank = 0
        self.remaining_indices = []
        self.num_samples = 0
        self.total_size = 0
        self.reset()

    def set_epoch(self, epoch):
        self.epoch = epoch
        self.processed_indices = set()
        self.reset()

    def record_batch(self, batch_idx, batch_size):
        """Record the dataset indices of this rank's `batch_idx` as
        processed (they are skipped after an elastic reset)."""
        shard = self.indices[batch_idx * batch_size:
                             (batch_idx + 1) * batch_size]
        self.processed_indices.update(shard)

    def reset(self):
        self.rank, self.num_replicas = _world()

        remaining = [idx for idx in range(len(self.dataset))
                     if idx not in self.processed_indices]
        if self.shuffle:
            random.Random(self.seed + self.epoch).shuffle(remaining)
        self.remaining_indices = remaining

        self.num_samples = int(math.ceil(
            len(self.remaining_indices) * 1.0 / max(self.num_replicas, 1)))
        self.total_size = self.num_samples * self.num_replicas

        indices = list(self.remaining_indices)
        # pad to make evenly divisible
        if indices:
            indices += indices[:(self.total_size - len(indices))]
        self.indices = indices[self.rank:self.total_size:self.num_replicas]

    def state_dict(self):
        return {
            "epoch": self.epoch,
            "processed_indices": sorted(self.processed_indices),
        }

    def load_state_dict(self, state_dict):
        self.epoch = state_dict["epoch"]
        self.processed_indices = set(state_dict["processed_indices"])

    def __iter__(self):
        return iter(self.indices)

    def __len__(self):
        return self.num_samples
