"""Complexity-balanced distributed samplers
(reference: bagua/torch_api/contrib/load_balancing_data_loader.py:12-325).

``LoadBalancingDistributedSampler`` sorts samples by a user complexity
function (e.g. sequence length), chunks the sorted order into
world-size-sized groups — so every rank gets a sample of comparable
cost per step — then shuffles whole chunks. Optional ``random_level``
jitters complexities to decorrelate epochs.

``LoadBalancingDistributedBatchSampler`` additionally lets a user
``batch_fn`` build variable-sized batches per rank, padding batch counts
so all ranks run the same number of steps.
"""

import math
from typing import Callable, Iterator, List, Optional

import torch
import torch.distributed as dist
from torch.utils.data.dataset import Dataset
from torch.utils.data.sampler import Sampler

__all__ = ["LoadBalancingDistributedSampler",
           "LoadBalancingDistributedBatchSampler"]


class LoadBalancingDistributedSampler(Sampler):
    def __init__(self, dataset: Dataset, complexity_fn: Callable,
                 num_replicas: Optional[int] = None,
                 rank: Optional[int] = None, shuffle: bool = True,
                 seed: int = 0, drop_last: bool = False,
                 random_level: float = 0.0):
        if num_replicas is None:
            if not dist.is_available() or not dist.is_initialized():
                raise RuntimeError("requires torch.distributed or explicit "
                                   "num_replicas")
            num_replicas = dist.get_world_size()
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        if rank >= num_replicas or rank < 0:
            raise ValueError("invalid rank %d for %d replicas"
                             % (rank, num_replicas))
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.epoch = 0
        self.drop_last = drop_last
        n = len(dataset)
        if self.drop_last and n % num_replicas != 0:
            self.num_samples = math.ceil((n - num_replicas) / num_replicas)
        else:
            self.num_samples = math.ceil(n / num_replicas)
        self.total_size = self.num_samples * num_replicas
        self.shuffle = shuffle
        self.seed = seed

        self.item_complexity = {i: complexity_fn(dataset[i])
                                for i in range(n)}
        self._sorted_indices = [
            k for k, _ in sorted(self.item_complexity.items(),
                                 key=lambda kv: kv[1])]
        cmax = max(self.item_complexity.values())
        cmin = min(self.item_complexity.values())
        if not 0.0 <= random_level <= 1.0:
            raise ValueError("random_level must be in [0, 1]")
        self.random_number = int((cmax - cmin) * random_level + 1)

    # ------------------------------------------------------------------
    def _chunks(self, order: List[int]) -> List[List[int]]:
        """Wrap-padded num_replicas-sized chunks of the sorted order."""
        num_chunks = max(1, self.num_samples)
        chunks, cur = [], []
        for i in range(num_chunks * self.num_replicas):
            cur.append(order[i % len(order)])
            if len(cur) == self.num_replicas:
                chunks.append(cur)
                cur = []
        return chunks

    def shuffle_chunks(self):
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            if self.random_number > 0:
                jitter = torch.randint(self.random_number,
                                       (len(self.item_complexity),),
                                       generator=g).tolist()
                jittered = {k: v + jitter[i] for i, (k, v) in
                            enumerate(self.item_complexity.items())}
                order = [k for k, _ in sorted(jittered.items(),
                                              key=lambda kv: kv[1])]
            else:
                order = self._sorted_indices
            index_chunks = self._chunks(order)
            chunk_indices = torch.randperm(len(index_chunks),
                                           generator=g).tolist()
        else:
            index_chunks = self._chunks(self._sorted_indices)
            chunk_indices = list(range(len(index_chunks)))

        if not self.drop_last:
            pad = self.num_samples - len(chunk_indices)
            if pad > 0:
                reps = math.ceil(pad / len(chunk_indices))
                chunk_indices += (chunk_indices * reps)[:pad]
        else:
            chunk_indices = chunk_indices[:self.num_samples]
        assert len(chunk_indices) == self.num_samples
        return index_chunks, chunk_indices

    def __iter__(self) -> Iterator[int]:
        index_chunks, chunk_indices = self.shuffle_chunks()
        return iter([index_chunks[i][self.rank] for i in chunk_indices])

    def __len__(self) -> int:
        return self.num_samples

    def set_epoch(self, epoch: int):
        self.epoch = epoch


class LoadBalancingDistributedBatchSampler(Sampler):
    def __init__(self, sampler: LoadBalancingDistributedSampler,
                 batch_fn: Callable, drop_last: bool = False):
        if not isinstance(sampler, LoadBalancingDistributedSampler):
            raise ValueError("sampler must be a "
                             "LoadBalancingDistributedSampler")
        if sampler.drop_last:
            raise ValueError("drop_last of the inner sampler must be False")
        self.sampler = sampler
        self.batch_fn = batch_fn
        self.drop_last = drop_last
        self.num_replicas = sampler.num_replicas
        self.rank = sampler.rank
        self.generate_batches()

    def generate_batches(self):
        index_chunks, chunk_indices = self.sampler.shuffle_chunks()
        batches = []
        for rank in range(self.num_replicas):
            sub = [index_chunks[i][rank] for i in chunk_indices]
            batches.append(self.batch_fn(sub))
        self.total_batch = (min(len(b) for b in batches) if self.drop_last
                            else max(len(b) for b in batches))
        self.padded_batches = [
            b + b[:self.total_batch - len(b)] if len(b) < self.total_batch
            else b[:self.total_batch]
            for b in batches]

    def __iter__(self):
        return iter(self.padded_batches[self.rank])

    def __len__(self):
        return self.total_batch

    def set_epoch(self, epoch: int):
        self.sampler.set_epoch(epoch)
        self.generate_batches()
