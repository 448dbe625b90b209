"""Train a MAML/MAML++ few-shot system (reference-compatible entry point).

Single GPU / CPU:
    python train_maml_system.py --name_of_args_json_file experiment_config/<cfg>.json

8x MI355X (task-level data parallelism over RCCL/xGMI):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 \
        train_maml_system.py --name_of_args_json_file <cfg>.json

Reference: ``train_maml_system.py:8-15`` builds args -> model -> data ->
ExperimentBuilder and runs it; this does the same with the distributed
context added.
"""

from howtotrainyourmamlpytorch_amd.config import get_args, select_device
from howtotrainyourmamlpytorch_amd.data import (MetaLearningSystemDataLoader,
                                                SyntheticEpisodeStream,
                                                maybe_unzip_dataset)
from howtotrainyourmamlpytorch_amd.experiment.builder import ExperimentBuilder
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier
from howtotrainyourmamlpytorch_amd.parallel import init_distributed


def main() -> None:
    args = get_args()
    device = select_device(args)
    dist_ctx = init_distributed(getattr(args, "distributed_backend", "auto"))

    model = MAMLFewShotClassifier(
        im_shape=(2, args.image_channels, args.image_height, args.image_width),
        device=device, args=args)
    model.attach_distributed(dist_ctx)

    if getattr(args, "synthetic_data", False):
        data = SyntheticEpisodeStream(args, rank=dist_ctx.rank,
                                      world_size=dist_ctx.world_size)
    else:
        maybe_unzip_dataset(args)
        data = MetaLearningSystemDataLoader(args, rank=dist_ctx.rank,
                                            world_size=dist_ctx.world_size)

    builder = ExperimentBuilder(args=args, data=data, model=model, device=device,
                                dist_ctx=dist_ctx)
    builder.run_experiment()


if __name__ == "__main__":
    main()
