"""End-to-end CLI smoke: the reference-compatible entry point runs a full
tiny experiment from a JSON config (CPU, synthetic data)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_train_maml_system_cli(tmp_path):
    cfg = {
        "experiment_name": "cli_smoke",
        "experiment_root": str(tmp_path),
        "dataset_name": "synthetic_omniglot",
        "synthetic_data": True,
        "batch_size": 2,
        "num_classes_per_set": 3,
        "num_samples_per_class": 1,
        "num_target_samples": 1,
        "image_height": 14, "image_width": 14, "image_channels": 1,
        "cnn_num_filters": 4, "num_stages": 3,
        "number_of_training_steps_per_iter": 2,
        "number_of_evaluation_steps_per_iter": 2,
        "total_epochs": 1, "total_iter_per_epoch": 2,
        "num_evaluation_tasks": 2,
        "max_models_to_save": 1,
        "seed": 0,
    }
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    env = dict(os.environ, MAML355_NO_HIP="1")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "train_maml_system.py"),
         "--name_of_args_json_file", str(cfg_path)],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    base = tmp_path / "cli_smoke"
    assert (base / "saved_models" / "train_model_latest").is_file()
    assert (base / "logs" / "test_summary.csv").is_file()

    # evaluate-only mode reuses the stored checkpoints
    cfg["evaluate_on_test_set_only"] = True
    cfg_path.write_text(json.dumps(cfg))
    out2 = subprocess.run(
        [sys.executable, os.path.join(REPO, "train_maml_system.py"),
         "--name_of_args_json_file", str(cfg_path)],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert out2.returncode == 0, out2.stderr[-2000:]
    assert "ensemble" in out2.stdout