"""End-to-end experiment tests on CPU with synthetic data: full epoch loop,
checkpoint/resume, pause, ensemble test, CSV/JSON artifacts."""

import os

import pytest
import torch

from howtotrainyourmamlpytorch_amd.config import get_args
from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream
from howtotrainyourmamlpytorch_amd.experiment.builder import ExperimentBuilder
from howtotrainyourmamlpytorch_amd.experiment.storage import load_statistics
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier


def exp_args(tmp_path, name="exp", **over):
    args = get_args([
        "--experiment_name", name,
        "--experiment_root", str(tmp_path),
        "--dataset_name", "synthetic_omniglot",
        "--batch_size", "2",
        "--num_classes_per_set", "3",
        "--num_samples_per_class", "1",
        "--num_target_samples", "1",
        "--image_height", "14", "--image_width", "14", "--image_channels", "1",
        "--cnn_num_filters", "4", "--num_stages", "3",
        "--number_of_training_steps_per_iter", "2",
        "--number_of_evaluation_steps_per_iter", "2",
        "--total_epochs", "2", "--total_iter_per_epoch", "2",
        "--num_evaluation_tasks", "4",
        "--max_models_to_save", "2",
        "--seed", "1",
        "--synthetic_data", "True",
    ])
    for k, v in over.items():
        setattr(args, k, v)
    return args


def build(args):
    device = torch.device("cpu")
    model = MAMLFewShotClassifier(
        im_shape=(2, args.image_channels, args.image_height, args.image_width),
        device=device, args=args)
    data = SyntheticEpisodeStream(args)
    return ExperimentBuilder(args=args, data=data, model=model, device=device)


def test_full_experiment_runs_and_writes_artifacts(tmp_path):
    args = exp_args(tmp_path)
    builder = build(args)
    builder.run_experiment()
    base = tmp_path / "exp"
    assert (base / "saved_models" / "train_model_0").is_file()
    assert (base / "saved_models" / "train_model_1").is_file()
    assert (base / "saved_models" / "train_model_latest").is_file()
    stats = load_statistics(str(base / "logs"))
    assert "val_accuracy_mean" in stats and len(stats["val_accuracy_mean"]) == 2
    assert (base / "logs" / "summary_statistics.json").is_file()
    assert (base / "logs" / "test_summary.csv").is_file()
    test_stats = load_statistics(str(base / "logs"), filename="test_summary.csv")
    acc = float(test_stats["test_accuracy_mean"][0])
    assert 0.0 <= acc <= 1.0


def test_pause_and_resume_continues_task_stream(tmp_path):
    # Run 1: pause after 1 epoch
    args = exp_args(tmp_path, name="resume_exp", total_epochs_before_pause=1)
    builder = build(args)
    with pytest.raises(SystemExit):
        builder.run_experiment()
    assert builder.state["current_iter"] == 2

    # Run 2: resume from latest and finish
    args2 = exp_args(tmp_path, name="resume_exp")
    builder2 = build(args2)
    assert builder2.state["current_iter"] == 2
    assert builder2.start_epoch == 1
    builder2.run_experiment()
    assert builder2.state["current_iter"] == 4

    # compare against an uninterrupted run: final theta must match exactly
    args3 = exp_args(tmp_path, name="oneshot_exp")
    builder3 = build(args3)
    builder3.run_experiment()
    torch.testing.assert_close(builder2.model.classifier.theta,
                               builder3.model.classifier.theta)
    torch.testing.assert_close(builder2.model.inner_loop_lrs,
                               builder3.model.inner_loop_lrs)


def test_checkpoint_roundtrip_bitwise(tmp_path):
    args = exp_args(tmp_path, name="ckpt_exp")
    builder = build(args)
    builder.run_experiment()
    model = builder.model
    save_dir = str(tmp_path / "ckpt_exp" / "saved_models")
    state = model.load_model(save_dir, "train_model", "latest")
    assert state["current_iter"] == 4
    sd_latest = {k: v.clone() for k, v in model.state_dict().items()}
    # perturb by loading a different checkpoint, then restore latest
    model.load_model(save_dir, "train_model", 0)
    assert any(not torch.equal(v, sd_latest[k])
               for k, v in model.state_dict().items())
    model.load_model(save_dir, "train_model", "latest")
    for k, v in model.state_dict().items():
        torch.testing.assert_close(v, sd_latest[k], rtol=0, atol=0)


def test_from_scratch_ignores_existing_checkpoint(tmp_path):
    args = exp_args(tmp_path, name="scratch_exp", total_epochs_before_pause=1)
    with pytest.raises(SystemExit):
        build(args).run_experiment()
    args2 = exp_args(tmp_path, name="scratch_exp", continue_from_epoch="from_scratch")
    builder2 = build(args2)
    assert builder2.state["current_iter"] == 0


def test_full_experiment_on_real_omniglot_npz(tmp_path):
    """Builder + real episode loader (shipped npz) + checkpointing +
    ensemble test, end to end on CPU with a tiny model."""
    args = get_args([
        "--experiment_name", "omni_cpu_exp",
        "--experiment_root", str(tmp_path),
        "--dataset_name", "omniglot_dataset",
        "--dataset_path", "datasets/omniglot_28x28.npz",
        "--batch_size", "2",
        "--num_classes_per_set", "3",
        "--num_samples_per_class", "1",
        "--num_target_samples", "1",
        "--image_height", "28", "--image_width", "28", "--image_channels", "1",
        "--cnn_num_filters", "4", "--num_stages", "3",
        "--number_of_training_steps_per_iter", "1",
        "--number_of_evaluation_steps_per_iter", "1",
        "--total_epochs", "1", "--total_iter_per_epoch", "2",
        "--num_evaluation_tasks", "2",
        "--max_models_to_save", "1",
        "--num_dataprovider_workers", "0",
        "--seed", "2",
    ])
    from howtotrainyourmamlpytorch_amd.data import MetaLearningSystemDataLoader
    device = torch.device("cpu")
    model = MAMLFewShotClassifier(im_shape=(2, 1, 28, 28), device=device, args=args)
    data = MetaLearningSystemDataLoader(args)
    builder = ExperimentBuilder(args=args, data=data, model=model, device=device)
    builder.run_experiment()
    base = tmp_path / "omni_cpu_exp"
    assert (base / "saved_models" / "train_model_latest").is_file()
    assert (base / "logs" / "test_summary.csv").is_file()


def test_resume_into_fresh_logs_dir_writes_csv_header(tmp_path):
    """Resuming after epoch 0 into a logs dir whose CSV is missing must
    re-create the header (header keyed on file existence, not epoch 0)."""
    args = exp_args(tmp_path, name="hdr", total_epochs_before_pause=1)
    builder = build(args)
    with pytest.raises(SystemExit):
        builder.run_experiment()
    csv_path = tmp_path / "hdr" / "logs" / "summary_statistics.csv"
    assert csv_path.is_file()
    os.remove(csv_path)  # simulate deleted/crashed logs
    args2 = exp_args(tmp_path, name="hdr")
    builder2 = build(args2)
    builder2.run_experiment()
    stats = load_statistics(str(tmp_path / "hdr" / "logs"))
    assert "val_accuracy_mean" in stats          # first row is a header
    assert len(stats["val_accuracy_mean"]) == 1  # one epoch-1 data row


def test_inner_loop_bn_params_shape_matches_reference(tmp_path):
    """enable_inner_loop_optimizable_bn_params: BN gamma/beta fast weights
    are a single [num_features] tensor even under per_step_bn_statistics
    (reference MetaBatchNormLayer override,
    meta_neural_network_architectures.py:194-198)."""
    args = exp_args(tmp_path, name="bnp",
                    enable_inner_loop_optimizable_bn_params=True)
    model = MAMLFewShotClassifier(
        im_shape=(2, args.image_channels, args.image_height, args.image_width),
        device=torch.device("cpu"), args=args)
    sd = model.reference_state_dict()
    w = sd["classifier.layer_dict.conv0.norm_layer.weight"]
    assert tuple(w.shape) == (args.cnn_num_filters,)
    # and the model trains with them as fast weights
    from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream
    batch = next(iter(SyntheticEpisodeStream(args).get_train_batches(1)))
    losses, _ = model.run_train_iter(batch, epoch=0)
    assert "loss" in losses
