"""Dataset unpack helpers (reference: ``utils/dataset_tools.py``): extract
``<dataset>.tar.bz2`` into ``$DATASET_DIR`` with file-count validation and
re-extract when the count is wrong (Omniglot 1623 classes x 20 samples,
mini-imagenet 100 x 600)."""

from __future__ import annotations

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
