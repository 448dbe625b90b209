"""Synthetic episode stream — benchmark/CI data source.

Produces batches with exactly the shapes and dtypes of the real pipeline
(``[B, N, S, c, h, w]`` images, int64 labels), seeded per global task index
with the same seed arithmetic as the real dataset, but without touching the
filesystem.  Used by ``bench.py`` (no network for datasets) and by tests.
"""

from __future__ import annotations

import torch


class SyntheticEpisodeStream:
    """``structured=False`` (default): pure random pixels — right shapes and
    bandwidth for benchmarking.  ``structured=True``: a *learnable* few-shot
    task distribution — ``num_latent_classes`` fixed random prototypes;
    episodes sample N of them and emit prototype+noise images with
    consistent labels, so meta-training must raise target accuracy above
    chance (used by the GPU learning test)."""

    def __init__(self, args, rank: int = 0, world_size: int = 1,
                 structured: bool = False, num_latent_classes: int = 64,
                 noise: float = 0.35):
        self.args = args
        self.rank = rank
        self.world_size = world_size
        self.global_batch = args.batch_size
        if self.global_batch % world_size != 0:
            raise ValueError("batch_size must divide world_size")
        self.local_batch = self.global_batch // world_size
        self.total_train_iters_produced = 0
        self.structured = structured
        self.noise = noise
        if structured:
            g = torch.Generator().manual_seed(1234 + args.seed)
            c, h, w = args.image_channels, args.image_height, args.image_width
            self.prototypes = torch.randn(num_latent_classes, c, h, w, generator=g)

    def continue_from_iter(self, current_iter: int) -> None:
        self.total_train_iters_produced = current_iter

    def _episode(self, seed: int):
        a = self.args
        g = torch.Generator().manual_seed(seed & 0x7FFFFFFF)
        n, s, t = a.num_classes_per_set, a.num_samples_per_class, a.num_target_samples
        c, h, w = a.image_channels, a.image_height, a.image_width
        if self.structured:
            cls = torch.randperm(self.prototypes.shape[0], generator=g)[:n]
            proto = self.prototypes[cls]                       # [n, c, h, w]
            xs = proto.unsqueeze(1) + self.noise * torch.randn(
                n, s, c, h, w, generator=g)
            xt = proto.unsqueeze(1) + self.noise * torch.randn(
                n, t, c, h, w, generator=g)
        else:
            xs = torch.rand(n, s, c, h, w, generator=g)
            xt = torch.rand(n, t, c, h, w, generator=g)
        ys = torch.arange(n).view(n, 1).expand(n, s).contiguous()
        yt = torch.arange(n).view(n, 1).expand(n, t).contiguous()
        return xs, xt, ys, yt

    def _batch(self, set_name: str, it: int):
        base_seed = {"train": self.args.train_seed, "val": self.args.val_seed,
                     "test": self.args.val_seed}[set_name]
        base = it * self.global_batch + self.rank * self.local_batch
        eps = [self._episode(base_seed + base + j) for j in range(self.local_batch)]
        return tuple(torch.stack([e[k] for e in eps]) for k in range(4))

    def get_train_batches(self, total_batches: int, augment_images: bool = False):
        start = self.total_train_iters_produced
        self.total_train_iters_produced += total_batches
        for i in range(start, start + total_batches):
            yield self._batch("train", i)

    def get_val_batches(self, total_batches: int, augment_images: bool = False):
        for i in range(total_batches):
            yield self._batch("val", i)

    def get_test_batches(self, total_batches: int, augment_images: bool = False):
        for i in range(total_batches):
            yield self._batch("test", i)
