"""Batch iterator over task episodes (reference:
``MetaLearningSystemDataLoader``, ``data.py:555-637``) with rank-aware
sharding.

Determinism contract: the episode for global task index ``g`` has seed
``set_seed + g`` regardless of world size.  For iteration ``i``, rank ``r``
of ``W`` ranks owns global indices ``i*GB + r*LB + j`` (``GB`` global
meta-batch, ``LB = GB/W``), so an 8-GPU run consumes exactly the same task
stream as a 1-GPU run — results are invariant to GPU count modulo
floating-point reduction order.
"""

from __future__ import annotations

from typing import Iterator

import torch

from .episodes import FewShotEpisodeDataset


class _ShardSampler(torch.utils.data.Sampler):
    def __init__(self, num_batches: int, global_batch: int, local_batch: int,
                 rank: int, start_iter: int = 0):
        self.num_batches = num_batches
        self.global_batch = global_batch
        self.local_batch = local_batch
        self.rank = rank
        self.start_iter = start_iter

    def __iter__(self) -> Iterator[int]:
        for i in range(self.start_iter, self.start_iter + self.num_batches):
            base = i * self.global_batch + self.rank * self.local_batch
            for j in range(self.local_batch):
                yield base + j

    def __len__(self) -> int:
        return self.num_batches * self.local_batch


class MetaLearningSystemDataLoader:
    def __init__(self, args, current_iter: int = 0, rank: int = 0, world_size: int = 1):
        self.args = args
        self.rank = rank
        self.world_size = world_size
        # samples_per_iter multiplies the tasks consumed per iteration,
        # exactly like the reference's DataLoader batch_size multiplier
        # (data.py:575-581); num_of_gpus is NOT multiplied in — GPU count
        # is carried by world_size sharding instead.
        self.global_batch = args.batch_size * int(getattr(args, "samples_per_iter", 1) or 1)
        if self.global_batch % world_size != 0:
            raise ValueError(f"batch_size {self.global_batch} not divisible by "
                             f"world_size {world_size}")
        self.local_batch = self.global_batch // world_size
        self.num_workers = getattr(args, "num_dataprovider_workers", 0)
        self.dataset = FewShotEpisodeDataset(args, current_set="train")
        self.total_train_iters_produced = current_iter

    def continue_from_iter(self, current_iter: int) -> None:
        """Resume the train task stream (reference: ``data.py:583-588``)."""
        self.total_train_iters_produced = current_iter

    def _make_loader(self, set_name: str, num_batches: int, augment: bool,
                     start_iter: int = 0):
        self.dataset.switch_set(set_name, current_iter=None)
        self.dataset.augment_images = augment
        sampler = _ShardSampler(num_batches, self.global_batch, self.local_batch,
                                self.rank, start_iter=start_iter)
        return torch.utils.data.DataLoader(
            self.dataset, batch_size=self.local_batch, sampler=sampler,
            num_workers=self.num_workers, drop_last=True,
            persistent_workers=False)

    def get_train_batches(self, total_batches: int, augment_images: bool = False):
        """Yields ``total_batches`` train batches starting at the resumable
        stream position.  Seeds advance by iteration count (reference
        rebases the seed, ``data.py:536-542,590-604``; here the global task
        index does the same job)."""
        start = self.total_train_iters_produced
        self.total_train_iters_produced += total_batches
        loader = self._make_loader("train", total_batches, augment_images,
                                   start_iter=start)
        for batch in loader:
            yield batch

    def get_val_batches(self, total_batches: int, augment_images: bool = False):
        loader = self._make_loader("val", total_batches, augment_images)
        for batch in loader:
            yield batch

    def get_test_batches(self, total_batches: int, augment_images: bool = False):
        loader = self._make_loader("test", total_batches, augment_images)
        for batch in loader:
            yield batch
