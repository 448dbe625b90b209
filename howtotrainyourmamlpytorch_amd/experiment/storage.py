"""Experiment directory layout + CSV/JSON statistics persistence.

Same on-disk contract as the reference (``utils/storage.py``): an
experiment folder containing ``saved_models/``, ``logs/`` and
``visual_outputs/``; per-epoch rows appended to
``logs/summary_statistics.csv``; cumulative JSON summaries; test results in
``logs/test_summary.csv``.
"""

from __future__ import annotations

import csv
import json
import os
from typing import Any, Dict, List, Sequence, Tuple


def build_experiment_folder(experiment_name: str, root: str = ".") -> Tuple[str, str, str]:
    """Create ``<root>/<experiment_name>/{saved_models,logs,visual_outputs}``
    and return (saved_models, logs, visual_outputs) paths
    (reference: ``utils/storage.py:49-66``)."""
    base = os.path.join(os.path.abspath(root), experiment_name)
    saved_models = os.path.join(base, "saved_models")
    logs = os.path.join(base, "logs")
    visual = os.path.join(base, "visual_outputs")
    for d in (base, saved_models, logs, visual):
        os.makedirs(d, exist_ok=True)
    return saved_models, logs, visual


def save_statistics(log_dir: str, statistics: Sequence[Any],
                    filename: str = "summary_statistics.csv", create: bool = False) -> str:
    """Append one CSV row (or create the file with a header row when
    ``create``) — reference: ``utils/storage.py:18-29``."""
    path = os.path.join(log_dir, filename)
    mode = "w" if create else "a"
    with open(path, mode, newline="") as f:
        writer = csv.writer(f)
        writer.writerow(list(statistics))
    return path


def load_statistics(log_dir: str, filename: str = "summary_statistics.csv") -> Dict[str, List[str]]:
    """Load a stats CSV into a dict of column -> list of values
    (reference: ``utils/storage.py:31-46``)."""
    path = os.path.join(log_dir, filename)
    with open(path, "r", newline="") as f:
        rows = list(csv.reader(f))
    if not rows:
        return {}
    header, body = rows[0], rows[1:]
    return {col: [row[i] if i < len(row) else "" for row in body]
            for i, col in enumerate(header)}


def save_to_json(filename: str, dict_to_store: Dict[str, Any]) -> None:
    with open(filename, "w") as f:
        json.dump(dict_to_store, f, indent=2, sort_keys=True, default=str)


def load_from_json(filename: str) -> Dict[str, Any]:
    with open(filename, "r") as f:
        return json.load(f)


# ---------------------------------------------------------------------------
# experiment-log JSON helpers (reference: ``utils/storage.py:82-128`` —
# defined there but never called by the reference's own builder; kept for
# API-surface parity and usable programmatically).
# ---------------------------------------------------------------------------
def create_json_experiment_log(experiment_log_dir: str, args,
                               log_name: str = "experiment_log.json") -> str:
    import datetime
    path = os.path.join(experiment_log_dir, log_name)
    summary = {k: v for k, v in vars(args).items()
               if isinstance(v, (int, float, str, bool, list, type(None)))}
    ts = datetime.datetime.now().timestamp()
    summary["epoch_stats"] = {}
    summary["experiment_status"] = [(ts, "initialization")]
    summary["experiment_initialization_time"] = ts
    with open(path, "w") as f:
        json.dump(summary, f)
    return path


def update_json_experiment_log_dict(key: str, value, experiment_log_dir: str,
                                    log_name: str = "experiment_log.json") -> None:
    path = os.path.join(experiment_log_dir, log_name)
    with open(path) as f:
        summary = json.load(f)
    summary[key].append(value)
    with open(path, "w") as f:
        json.dump(summary, f)


def change_json_log_experiment_status(experiment_status: str,
                                      experiment_log_dir: str,
                                      log_name: str = "experiment_log.json") -> None:
    import datetime
    update_json_experiment_log_dict(
        "experiment_status",
        (datetime.datetime.now().timestamp(), experiment_status),
        experiment_log_dir, log_name)


def update_json_experiment_log_epoch_stats(epoch_stats: Dict[str, Any],
                                           experiment_log_dir: str,
                                           log_name: str = "experiment_log.json") -> str:
    path = os.path.join(experiment_log_dir, log_name)
    with open(path) as f:
        summary = json.load(f)
    stats = summary["epoch_stats"]
    for k, v in epoch_stats.items():
        stats.setdefault(k, []).append(float(v))
    with open(path, "w") as f:
        json.dump(summary, f)
    return path


def get_best_validation_model_statistics(log_dir: str,
                                         filename: str = "summary_statistics.csv",
                                         key: str = "val_accuracy_mean"):
    """Best validation value and its epoch from the stats CSV (reference:
    ``utils/storage.py:68-80``; generalized to maximize accuracy keys and
    minimize loss keys)."""
    import numpy as np
    stats = load_statistics(log_dir, filename)
    vals = np.array([float(v) for v in stats[key]], dtype=np.float64)
    if "loss" in key:
        idx = int(np.argmin(vals))
    else:
        idx = int(np.argmax(vals))
    return float(vals[idx]), idx
