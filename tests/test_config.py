import json
import os

from howtotrainyourmamlpytorch_amd.config import get_args, finalize_args, build_parser


def test_defaults_parse():
    args = get_args([])
    assert args.batch_size == 16
    assert args.num_classes_per_set == 5
    assert args.second_order is True
    assert args.per_step_bn_statistics is True
    assert args.init_inner_loop_learning_rate == args.task_learning_rate


def test_json_overrides_and_bool_coercion(tmp_path):
    cfg = {
        "batch_size": 8,
        "num_classes_per_set": 20,
        "second_order": "true",
        "max_pooling": "false",
        "init_inner_loop_learning_rate": 0.05,
        "continue_from_epoch": 7,          # protected: must NOT override CLI
        "a_key_argparse_never_defined": 42,  # JSON may introduce new keys
    }
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps(cfg))
    args = get_args(["--name_of_args_json_file", str(p), "--continue_from_epoch", "latest"])
    assert args.batch_size == 8
    assert args.num_classes_per_set == 20
    assert args.second_order is True
    assert args.max_pooling is False
    assert args.init_inner_loop_learning_rate == 0.05
    assert args.continue_from_epoch == "latest"
    assert args.a_key_argparse_never_defined == 42


def test_reference_config_files_load():
    """Every shipped reference config must parse through our config system
    (JSON-schema compatibility)."""
    ref_dir = "/root/reference/experiment_config"
    if not os.path.isdir(ref_dir):
        return
    files = sorted(os.listdir(ref_dir))[:6]
    for fname in files:
        args = get_args(["--name_of_args_json_file", os.path.join(ref_dir, fname)])
        assert args.total_epochs == 100
        assert isinstance(args.second_order, bool)
        assert isinstance(args.per_step_bn_statistics, bool)
        assert args.num_stages == 4


def test_dataset_dir_rooting(tmp_path, monkeypatch):
    monkeypatch.setenv("DATASET_DIR", str(tmp_path))
    args = get_args(["--dataset_path", "omniglot_dataset"])
    assert args.dataset_path == str(tmp_path / "omniglot_dataset")
