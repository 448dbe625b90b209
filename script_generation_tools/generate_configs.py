"""Generate the full experiment-config matrix (reference:
``script_generation_tools/generate_configs.py``).

Fills ``$var$`` placeholders in ``experiment_template_config/*.json`` with
a hyperparameter grid x seeds {0,1,2}, writing one concrete JSON per cell
into ``experiment_config/`` — the same 36-experiment matrix the reference
ships (MAML vs MAML++ differ only in the three MAML++ switches).

Run: python script_generation_tools/generate_configs.py
"""

from __future__ import annotations

import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
TEMPLATE_DIR = os.path.join(REPO, "experiment_template_config")
OUTPUT_DIR = os.path.join(REPO, "experiment_config")

SEEDS = [0, 1, 2]

# (num_samples_per_class, batch_size, inner_lr, filters, ways)
GRIDS = {
    "omniglot": [
        (1, 8, 0.1, 64, 5),
        (5, 8, 0.1, 64, 5),
        (1, 8, 0.1, 64, 20),
        (5, 8, 0.1, 64, 20),
    ],
    "mini-imagenet": [
        (1, 2, 0.01, 48, 5),
        (5, 2, 0.01, 48, 5),
    ],
}


def fill(template: str, values: dict) -> str:
    out = template
    for key, value in values.items():
        out = out.replace(f"${key}$", str(value))
    return out


def main() -> None:
    os.makedirs(OUTPUT_DIR, exist_ok=True)
    count = 0
    for fname in sorted(os.listdir(TEMPLATE_DIR)):
        if not fname.endswith(".json"):
            continue
        dataset = "mini-imagenet" if "imagenet" in fname else "omniglot"
        variant = fname[:-len(".json")]  # e.g. omniglot_maml++
        with open(os.path.join(TEMPLATE_DIR, fname)) as f:
            template = f.read()
        for (shots, batch, inner_lr, filters, ways) in GRIDS[dataset]:
            for seed in SEEDS:
                exp = f"{dataset}_{shots}_{batch}_{inner_lr}_{filters}_{ways}_{seed}"
                values = dict(
                    num_samples_per_class=shots, batch_size=batch,
                    init_inner_loop_learning_rate=inner_lr,
                    cnn_num_filters=filters, num_classes_per_set=ways,
                    train_seed=seed, val_seed=seed,
                    experiment_name=exp,
                )
                out = fill(template, values)
                out_name = f"{variant}-{exp}.json"
                with open(os.path.join(OUTPUT_DIR, out_name), "w") as f:
                    f.write(out)
                count += 1
    print(f"wrote {count} configs to {OUTPUT_DIR}")


if __name__ == "__main__":
    main()
