"""Deterministic seeding, matching the reference's derivation: the torch
seed is drawn from ``np.random.RandomState(seed)``
(``few_shot_learning_system.py:13-23``) so that checkpoint-compatible runs
see the same initialization stream."""

from __future__ import annotations

import numpy as np
import torch


def set_torch_seed(seed: int) -> torch.Generator:
    rng = np.random.RandomState(seed=seed)
    torch_seed = int(rng.randint(0, 999999))
    torch.manual_seed(seed=torch_seed)
    gen = torch.Generator()
    gen.manual_seed(torch_seed)
    return gen
