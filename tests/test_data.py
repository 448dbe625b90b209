"""Data pipeline tests: episode determinism, split, loader sharding
invariance, synthetic stream."""

import os

import numpy as np
import pytest
import torch

from howtotrainyourmamlpytorch_amd.config import get_args
from howtotrainyourmamlpytorch_amd.data import (FewShotEpisodeDataset,
                                                MetaLearningSystemDataLoader,
                                                SyntheticEpisodeStream)


@pytest.fixture(scope="module")
def tiny_dataset(tmp_path_factory):
    """12 classes x 6 images of 10x10 grayscale PNGs, class encoded by the
    last two path components (alphabet/character) like Omniglot."""
    from PIL import Image
    root = tmp_path_factory.mktemp("tinyset") / "tiny_dataset"
    rng = np.random.RandomState(0)
    for a in range(3):
        for c in range(4):
            d = root / f"alpha{a}" / f"char{c}"
            d.mkdir(parents=True)
            for i in range(6):
                arr = rng.randint(0, 255, size=(10, 10), dtype=np.uint8)
                Image.fromarray(arr, mode="L").save(d / f"{i}.png")
    return str(root)


def data_args(root, **over):
    args = get_args([
        "--dataset_name", "tiny_dataset",
        "--dataset_path", root,
        "--num_classes_per_set", "3",
        "--num_samples_per_class", "2",
        "--num_target_samples", "1",
        "--image_height", "10", "--image_width", "10", "--image_channels", "1",
        "--batch_size", "4",
        "--num_dataprovider_workers", "0",
        "--train_val_test_split", "0.5", "0.25", "0.25",
        "--total_epochs", "2", "--total_iter_per_epoch", "3",
    ])
    for k, v in over.items():
        setattr(args, k, v)
    return args


def test_scan_split_and_cache(tiny_dataset):
    args = data_args(tiny_dataset)
    ds = FewShotEpisodeDataset(args, current_set="train")
    assert len(ds.datasets["train"]) == 6
    assert len(ds.datasets["val"]) == 3
    assert len(ds.datasets["test"]) == 3
    assert os.path.isfile(tiny_dataset.rstrip("/") + "_path_cache.json")
    # cache reload path
    ds2 = FewShotEpisodeDataset(args, current_set="train")
    assert sorted(ds2.datasets["train"]) == sorted(ds.datasets["train"])


def test_episode_is_pure_function_of_seed(tiny_dataset):
    args = data_args(tiny_dataset)
    ds = FewShotEpisodeDataset(args, current_set="train")
    xs1, xt1, ys1, yt1, s1 = ds.get_set("train", seed=42, augment=True)
    xs2, xt2, ys2, yt2, s2 = ds.get_set("train", seed=42, augment=True)
    torch.testing.assert_close(xs1, xs2)
    torch.testing.assert_close(xt1, xt2)
    assert xs1.shape == (3, 2, 1, 10, 10)
    assert xt1.shape == (3, 1, 1, 10, 10)
    assert ys1.shape == (3, 2) and yt1.shape == (3, 1)
    xs3, *_ = ds.get_set("train", seed=43, augment=True)
    assert not torch.allclose(xs1, xs3)


def test_rotation_augmentation_changes_images(tiny_dataset):
    args = data_args(tiny_dataset)
    ds = FewShotEpisodeDataset(args, current_set="train")
    # find a seed whose rotations are nonzero
    for seed in range(20):
        rng = np.random.RandomState(seed)
        rng.choice(6, size=3, replace=False)
        if rng.randint(0, 4, size=3).any():
            break
    xs_aug, *_ = ds.get_set("train", seed=seed, augment=True)
    xs_plain, *_ = ds.get_set("train", seed=seed, augment=False)
    assert not torch.allclose(xs_aug, xs_plain)


def test_load_into_memory_matches_disk(tiny_dataset):
    args = data_args(tiny_dataset)
    ds_disk = FewShotEpisodeDataset(args, current_set="train")
    args_mem = data_args(tiny_dataset, load_into_memory=True)
    ds_mem = FewShotEpisodeDataset(args_mem, current_set="train")
    a, *_ = ds_disk.get_set("train", seed=5)
    b, *_ = ds_mem.get_set("train", seed=5)
    torch.testing.assert_close(a, b)


def test_loader_shard_invariance(tiny_dataset):
    """The union of rank shards at world_size=2 must equal the world_size=1
    task stream, batch by batch."""
    args = data_args(tiny_dataset)
    one = MetaLearningSystemDataLoader(args, rank=0, world_size=1)
    b1 = list(one.get_train_batches(total_batches=2))
    r0 = MetaLearningSystemDataLoader(args, rank=0, world_size=2)
    r1 = MetaLearningSystemDataLoader(args, rank=1, world_size=2)
    b0 = list(r0.get_train_batches(total_batches=2))
    b1r = list(r1.get_train_batches(total_batches=2))
    for i in range(2):
        merged = torch.cat([b0[i][0], b1r[i][0]], dim=0)
        torch.testing.assert_close(merged, b1[i][0])


def test_loader_resume_stream(tiny_dataset):
    args = data_args(tiny_dataset)
    loader = MetaLearningSystemDataLoader(args, rank=0, world_size=1)
    first_then_second = list(loader.get_train_batches(total_batches=2))
    fresh = MetaLearningSystemDataLoader(args, rank=0, world_size=1)
    fresh.continue_from_iter(1)
    resumed = list(fresh.get_train_batches(total_batches=1))
    torch.testing.assert_close(resumed[0][0], first_then_second[1][0])


def test_synthetic_stream_shapes_and_sharding():
    args = data_args("/nonexistent")  # synthetic never touches the path
    s1 = SyntheticEpisodeStream(args, rank=0, world_size=1)
    (xs, xt, ys, yt) = next(iter(s1.get_train_batches(1)))
    assert xs.shape == (4, 3, 2, 1, 10, 10)
    assert yt.dtype == torch.int64
    r0 = SyntheticEpisodeStream(args, rank=0, world_size=2)
    r1 = SyntheticEpisodeStream(args, rank=1, world_size=2)
    a = next(iter(r0.get_train_batches(1)))
    b = next(iter(r1.get_train_batches(1)))
    merged = torch.cat([a[0], b[0]], dim=0)
    torch.testing.assert_close(merged, xs)


def test_pre_split_dataset_layout(tmp_path):
    """sets_are_pre_split: train/ val/ test/ subdirectories with their own
    class folders (the mini-imagenet layout, reference data.py:234-268)."""
    from PIL import Image
    rng = np.random.RandomState(1)
    root = tmp_path / "presplit_ds"
    counts = {"train": 5, "val": 3, "test": 3}
    for split, ncls in counts.items():
        for c in range(ncls):
            d = root / split / f"group{c}" / f"cls{c}"
            d.mkdir(parents=True)
            for i in range(4):
                arr = rng.randint(0, 255, size=(10, 10), dtype=np.uint8)
                Image.fromarray(arr, mode="L").save(d / f"{i}.png")
    args = data_args(str(root), sets_are_pre_split=True)
    ds = FewShotEpisodeDataset(args, current_set="train")
    assert len(ds.datasets["train"]) == 5
    assert len(ds.datasets["val"]) == 3
    assert len(ds.datasets["test"]) == 3
    xs, xt, ys, yt, seed = ds.get_set("val", seed=3)
    assert xs.shape == (3, 2, 1, 10, 10)


def test_full_res_mini_imagenet_ingestion(tmp_path):
    """Full-resolution mini-imagenet-style ingestion: 84x84 RGB class
    folders in the pre-split train/val/test layout (reference
    data.py:374-395 loads RGB /255; configs use sets_are_pre_split)."""
    from PIL import Image
    rng = np.random.RandomState(7)
    root = tmp_path / "mini_imagenet_full_size"
    counts = {"train": 6, "val": 4, "test": 4}
    for split, ncls in counts.items():
        for c in range(ncls):
            d = root / split / f"n{split}{c:08d}"
            d.mkdir(parents=True)
            for i in range(5):
                arr = rng.randint(0, 255, size=(84, 84, 3), dtype=np.uint8)
                Image.fromarray(arr, mode="RGB").save(d / f"{i}.jpg")
    args = data_args(str(root), sets_are_pre_split=True,
                     dataset_name="mini_imagenet_full_size")
    args.image_height = 84
    args.image_width = 84
    args.image_channels = 3
    ds = FewShotEpisodeDataset(args, current_set="train")
    assert len(ds.datasets["train"]) == 6
    assert len(ds.datasets["val"]) == 4
    xs, xt, ys, yt, seed = ds.get_set("train", seed=11)
    assert xs.shape == (3, 2, 3, 84, 84)
    assert xt.shape == (3, 1, 3, 84, 84)
    # RGB images are /255 then channel-standardized (reference
    # data.py:389-395 + the imagenet Normalize transform) — uniform-noise
    # input lands within a few stds of zero, not raw uint8 range
    assert float(xs.abs().max()) < 6.0
    assert float(xs.std()) > 0.2  # not all-zero / constant


def test_samples_per_iter_multiplies_tasks(tiny_dataset):
    """samples_per_iter multiplies tasks per yielded batch, matching the
    reference DataLoader batch-size multiplier (data.py:575-581)."""
    args = data_args(tiny_dataset)
    args.samples_per_iter = 2
    loader = MetaLearningSystemDataLoader(args, current_iter=0)
    batch = next(iter(loader.get_train_batches(1)))
    assert batch[0].shape[0] == 8  # batch_size 4 x samples_per_iter 2


def test_export_label_maps(tmp_path):
    from howtotrainyourmamlpytorch_amd.data.tools import export_label_maps
    import json
    names = ["alpha/char1", "alpha/char2", "beta/char1"]
    p1, p2 = export_label_maps("toy_ds", names, str(tmp_path))
    with open(p1) as f:
        fwd = json.load(f)
    with open(p2) as f:
        rev = json.load(f)
    assert fwd == {"alpha/char1": 0, "alpha/char2": 1, "beta/char1": 2}
    assert rev == {"0": "alpha/char1", "1": "alpha/char2", "2": "beta/char1"}


def test_shipped_label_maps_match_npz():
    """The label-map JSON pair shipped beside the Omniglot npz covers every
    class in the npz (C13 completeness)."""
    import json
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    npz = os.path.join(repo, "datasets", "omniglot_28x28.npz")
    lmap = os.path.join(repo, "datasets", "label_name_to_map_omniglot_28x28.json")
    data = np.load(npz, allow_pickle=True)
    with open(lmap) as f:
        fwd = json.load(f)
    names = [str(n) for n in data["class_names"]]
    assert len(fwd) == len(names) == 1623
    assert all(fwd[n] == i for i, n in enumerate(names))


def test_cifar_transform_branch(tmp_path):
    """cifar datasets use crop/flip + classification mean/std normalize
    (reference data.py:81-90) and NOT the omniglot rotations."""
    from PIL import Image
    rng = np.random.RandomState(3)
    root = tmp_path / "cifar10_fs"
    for c in range(6):
        d = root / "group" / f"cls{c}"
        d.mkdir(parents=True)
        for i in range(5):
            arr = rng.randint(0, 255, size=(32, 32, 3), dtype=np.uint8)
            Image.fromarray(arr, mode="RGB").save(d / f"{i}.png")
    args = data_args(str(root), dataset_name="cifar10_fs")
    args.image_height = 32
    args.image_width = 32
    args.image_channels = 3
    args.classification_mean = [0.49, 0.48, 0.45]
    args.classification_std = [0.25, 0.24, 0.26]
    args.train_val_test_split = [0.5, 0.25, 0.25]
    ds = FewShotEpisodeDataset(args, current_set="train")
    xs_a, xt_a, *_ = ds.get_set("train", seed=5, augment=True)
    xs_b, xt_b, *_ = ds.get_set("train", seed=5, augment=True)
    torch.testing.assert_close(xs_a, xs_b)  # augment draws are seed-pure
    xs_n, *_ = ds.get_set("train", seed=5, augment=False)
    assert not torch.allclose(xs_a, xs_n)   # crop/flip changed pixels
    # normalized: values centered near 0 with the given std
    assert float(xs_n.mean().abs()) < 1.0 and float(xs_n.abs().max()) < 4.0
