"""Re-run the top-N-checkpoint ensemble test of a finished experiment,
optionally at a different compute dtype (e.g. fp32 eval of a bf16-trained
run).  Usage:

    python tools/ensemble_eval.py <config.json> <experiment_root> [fp32]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from howtotrainyourmamlpytorch_amd.config import get_args, select_device  # noqa: E402
from howtotrainyourmamlpytorch_amd.data import MetaLearningSystemDataLoader  # noqa: E402
from howtotrainyourmamlpytorch_amd.experiment.builder import ExperimentBuilder  # noqa: E402
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier  # noqa: E402


def main():
    cfg, root = sys.argv[1], sys.argv[2]
    dtype = sys.argv[3] if len(sys.argv) > 3 else "bf16"
    args = get_args(["--name_of_args_json_file", cfg])
    args.experiment_root = root
    args.compute_dtype = dtype
    if os.path.isfile(str(args.dataset_path).rstrip(os.sep) + ".npz"):
        args.load_from_npz_files = True
    device = select_device(args)
    model = MAMLFewShotClassifier(
        im_shape=(2, args.image_channels, args.image_height, args.image_width),
        device=device, args=args)
    data = MetaLearningSystemDataLoader(args)
    builder = ExperimentBuilder(args=args, data=data, model=model, device=device)
    assert builder.state["per_epoch_statistics"].get("val_accuracy_mean"), \
        "no per-epoch stats in the resumed state"
    result = builder.evaluate_test_set_using_the_best_models(
        top_n_models=getattr(args, "max_models_to_save", 5))
    print("ENSEMBLE_RESULT", dtype, result)


if __name__ == "__main__":
    main()
