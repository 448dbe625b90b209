"""Coverage for auxiliary paths: layer-norm backbone, no-maxpool (stride-2)
backbone, dataset tools, phase timers."""

import os
import tarfile

import numpy as np
import pytest
import torch

from howtotrainyourmamlpytorch_amd.config import get_args
from howtotrainyourmamlpytorch_amd.data.tools import (count_files,
                                                      maybe_unzip_dataset,
                                                      unzip_file)
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier
from howtotrainyourmamlpytorch_amd.utils.timers import PhaseTimers


def eng_args(**over):
    args = get_args([
        "--batch_size", "2", "--num_classes_per_set", "3",
        "--num_samples_per_class", "1", "--num_target_samples", "1",
        "--image_height", "14", "--image_width", "14", "--image_channels", "1",
        "--cnn_num_filters", "4", "--num_stages", "3",
        "--number_of_training_steps_per_iter", "2", "--seed", "1",
    ])
    for k, v in over.items():
        setattr(args, k, v)
    return args


def batch_for(args, seed=0):
    g = torch.Generator().manual_seed(seed)
    N, S, T = args.num_classes_per_set, args.num_samples_per_class, args.num_target_samples
    c, h, w = args.image_channels, args.image_height, args.image_width
    B = args.batch_size
    xs = torch.randn(B, N, S, c, h, w, generator=g)
    xt = torch.randn(B, N, T, c, h, w, generator=g)
    ys = torch.arange(N).view(1, N, 1).expand(B, N, S).contiguous()
    yt = torch.arange(N).view(1, N, 1).expand(B, N, T).contiguous()
    return xs, xt, ys, yt


def test_layer_norm_backbone_trains():
    args = eng_args(norm_layer="layer_norm")
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14),
                                  device=torch.device("cpu"), args=args)
    theta0 = model.classifier.theta.detach().clone()
    losses, _ = model.run_train_iter(batch_for(args), epoch=0)
    assert losses["loss"] > 0
    assert not torch.allclose(model.classifier.theta.detach(), theta0)


def test_stride2_no_maxpool_backbone_trains():
    args = eng_args(max_pooling=False)
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14),
                                  device=torch.device("cpu"), args=args)
    # no-maxpool path: global avg pool -> feature dim == num_filters
    assert model.classifier.feature_dim == 4
    losses, _ = model.run_train_iter(batch_for(args), epoch=0)
    assert losses["loss"] > 0


def test_inner_loop_bn_params_variant():
    args = eng_args(enable_inner_loop_optimizable_bn_params=True)
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14),
                                  device=torch.device("cpu"), args=args)
    names = model.classifier.arena.names()
    assert any("norm_layer.weight" in n for n in names)
    losses, _ = model.run_train_iter(batch_for(args), epoch=0)
    assert losses["loss"] > 0


def test_maybe_unzip_dataset(tmp_path):
    src = tmp_path / "fake_dataset"
    (src / "a" / "b").mkdir(parents=True)
    for i in range(3):
        (src / "a" / "b" / f"{i}.png").write_bytes(b"x")
    archive = tmp_path / "fake_dataset.tar.bz2"
    with tarfile.open(archive, "w:bz2") as tf:
        tf.add(src, arcname="fake_dataset")
    # remove the extracted dir; maybe_unzip must restore it
    import shutil
    shutil.rmtree(src)
    args = get_args(["--dataset_name", "fake_dataset",
                     "--dataset_path", str(src)])
    maybe_unzip_dataset(args)
    assert count_files(str(src)) == 3


def test_phase_timers_cpu():
    t = PhaseTimers(use_cuda_events=False)
    with t.phase("a"):
        sum(range(1000))
    with t.phase("a"):
        pass
    s = t.summary()
    assert s["a_count"] == 2
    assert s["a_ms"] >= 0


def test_engine_with_phase_timers():
    args = eng_args(enable_phase_timers=True)
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14),
                                  device=torch.device("cpu"), args=args)
    model.run_train_iter(batch_for(args), epoch=0)
    s = model.timers.summary()
    assert "inner_loop_fwd_ms" in s and "outer_bwd_and_opt_ms" in s


def test_slot_gather_gradcheck():
    """The deterministic LSLR gather (vgg._SlotGather) must be exact to
    autograd through second order (gradcheck on tiny doubles)."""
    import torch
    from howtotrainyourmamlpytorch_amd.models.vgg import _SlotGather
    bounds = [(0, 3), (3, 2), (5, 4)]
    slot_index = torch.tensor([0, 0, 0, 1, 1, 2, 2, 2, 2])
    lrs = torch.randn(3, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda l: _SlotGather.apply(l, slot_index, bounds), (lrs,))
    assert torch.autograd.gradgradcheck(
        lambda l: (_SlotGather.apply(l, slot_index, bounds) ** 2).sum(), (lrs,))


def test_chunked_train_step_losses_are_weighted_means():
    """The chunked path's losses dict must be the task-weighted mean over
    chunks for every scalar entry."""
    import torch
    from howtotrainyourmamlpytorch_amd.config import get_args
    from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream
    from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier
    args = get_args([
        "--batch_size", "4", "--num_classes_per_set", "3",
        "--num_samples_per_class", "1", "--num_target_samples", "1",
        "--image_height", "10", "--image_width", "10", "--image_channels", "1",
        "--cnn_num_filters", "4", "--num_stages", "2",
        "--number_of_training_steps_per_iter", "2",
        "--multi_step_loss_num_epochs", "10",
        "--total_epochs", "4", "--seed", "5", "--dataset_name", "synthetic",
    ])
    args.task_chunk_size = 2
    model = MAMLFewShotClassifier(im_shape=(2, 1, 10, 10),
                                  device=torch.device("cpu"), args=args)
    batch = next(iter(SyntheticEpisodeStream(args).get_train_batches(1)))
    losses, preds = model.run_train_iter(batch, epoch=0)
    assert isinstance(losses["loss"], float)
    assert 0.0 <= losses["accuracy"] <= 1.0
    assert preds.shape[0] == 4  # all tasks' predictions concatenated
    # MSL importance entries survive chunking and sum to ~1
    iv = [v for k, v in losses.items() if k.startswith("loss_importance_vector_")]
    assert iv and abs(sum(iv) - 1.0) < 1e-6


def test_experiment_log_json_helpers(tmp_path):
    from howtotrainyourmamlpytorch_amd.experiment import storage as st
    from howtotrainyourmamlpytorch_amd.config import get_args
    args = get_args(["--experiment_name", "x"])
    p = st.create_json_experiment_log(str(tmp_path), args)
    st.change_json_log_experiment_status("training", str(tmp_path))
    st.update_json_experiment_log_epoch_stats(
        {"val_accuracy_mean": 0.5, "val_loss_mean": 1.2}, str(tmp_path))
    st.update_json_experiment_log_epoch_stats(
        {"val_accuracy_mean": 0.8, "val_loss_mean": 0.7}, str(tmp_path))
    import json as js
    d = js.load(open(p))
    assert d["epoch_stats"]["val_accuracy_mean"] == [0.5, 0.8]
    assert d["experiment_status"][-1][1] == "training"
    # best-val helper over a real CSV
    st.save_statistics(str(tmp_path), ["epoch", "val_accuracy_mean"], create=True)
    st.save_statistics(str(tmp_path), [0, 0.4])
    st.save_statistics(str(tmp_path), [1, 0.9])
    st.save_statistics(str(tmp_path), [2, 0.6])
    best, epoch = st.get_best_validation_model_statistics(str(tmp_path))
    assert best == 0.9 and epoch == 1
