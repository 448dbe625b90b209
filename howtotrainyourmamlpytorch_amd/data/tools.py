"""Dataset unpack helpers (reference: ``utils/dataset_tools.py``): extract
``<dataset>.tar.bz2`` into ``$DATASET_DIR`` with file-count validation and
re-extract when the count is wrong (Omniglot 1623 classes x 20 samples,
mini-imagenet 100 x 600)."""

from __future__ import annotations

import json
import os
import shutil
import subprocess
import tarfile

EXPECTED_FILE_COUNTS = {
    "omniglot_dataset": 1623 * 20,
    "mini_imagenet": 100 * 600,
    "mini_imagenet_full_size": 100 * 600,
}


def count_files(root: str) -> int:
    total = 0
    for _dirpath, _dirs, files in os.walk(root):
        total += len(files)
    return total


def unzip_file(archive_path: str, dest_dir: str) -> None:
    os.makedirs(dest_dir, exist_ok=True)
    if shutil.which("pbzip2"):
        subprocess.run(["tar", "-I", "pbzip2", "-xf", archive_path, "-C", dest_dir],
                       check=True)
    else:
        with tarfile.open(archive_path, "r:bz2") as tf:
            tf.extractall(dest_dir)


def maybe_unzip_dataset(args) -> None:
    """Ensure ``args.dataset_path`` exists and contains the expected number
    of files; otherwise (re-)extract ``<dataset_path>.tar.bz2``."""
    path = args.dataset_path
    # compact-npz backend needs no extraction
    if str(path).endswith(".npz") and os.path.isfile(path):
        return
    if getattr(args, "load_from_npz_files", False) and \
            os.path.isfile(str(path).rstrip(os.sep) + ".npz"):
        return
    expected = EXPECTED_FILE_COUNTS.get(args.dataset_name)
    ok = os.path.isdir(path) and (expected is None or count_files(path) >= expected)
    if ok:
        return
    archive = path.rstrip(os.sep) + ".tar.bz2"
    if not os.path.isfile(archive):
        if os.path.isdir(path):
            return  # partial dataset, no archive to fix it with — let the scan proceed
        raise FileNotFoundError(
            f"dataset not found at {path} and no archive at {archive}")
    if os.path.isdir(path):
        shutil.rmtree(path)
    unzip_file(archive, os.path.dirname(os.path.abspath(path)) or ".")


def export_label_maps(dataset_name: str, class_names, out_dir: str) -> tuple:
    """Write the reference's label-map JSON pair next to a dataset
    (``datasets/label_name_to_map_<ds>.json`` mapping class name -> index
    and ``map_to_label_name_<ds>.json`` mapping index -> class name —
    reference files ``/root/reference/datasets/label_name_to_map_*.json``).
    Returns the two paths."""
    os.makedirs(out_dir, exist_ok=True)
    name_to_idx = {str(n): i for i, n in enumerate(class_names)}
    idx_to_name = {str(i): str(n) for i, n in enumerate(class_names)}
    p1 = os.path.join(out_dir, f"label_name_to_map_{dataset_name}.json")
    p2 = os.path.join(out_dir, f"map_to_label_name_{dataset_name}.json")
    with open(p1, "w") as f:
        json.dump(name_to_idx, f)
    with open(p2, "w") as f:
        json.dump(idx_to_name, f)
    return p1, p2
