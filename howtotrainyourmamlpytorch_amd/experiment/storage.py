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
