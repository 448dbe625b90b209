"""Flagship benchmark: MAML++ meta-training throughput (meta-tasks/sec).

Config = BASELINE.json's headline: Mini-ImageNet 5-way 1-shot MAML++
(48 filters, 84x84x3, 5 inner steps, LSLR+MSL, second-order), synthetic
episodes, random-init weights, bf16 conv compute / fp32 master.

Weak scaling: per-GPU meta-batch is fixed (default 128 tasks/GPU, sized
well within 288 GB HBM3E per BASELINE.json config #5); the
reported ``value`` is whole-job tasks/sec over all ranks.

Single GPU:   python bench.py --gpus 1 --steps 20 --warmup 5
Multi-GPU (driver does this):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

from howtotrainyourmamlpytorch_amd.config import get_args, select_device
from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier
from howtotrainyourmamlpytorch_amd.parallel import init_distributed


def parse_cli():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--tasks_per_gpu", type=int, default=128)
    p.add_argument("--inner_steps", type=int, default=5)
    p.add_argument("--model", type=str, default="maml++_miniimagenet_5w1s",
                   choices=["maml++_miniimagenet_5w1s", "maml++_omniglot_20w5s"])
    p.add_argument("--second_order", type=str, default="True")
    return p.parse_args()


def build_args(cli, world_size: int):
    if cli.model == "maml++_miniimagenet_5w1s":
        overrides = dict(dataset_name="mini_imagenet_full_size", image_height=84,
                         image_width=84, image_channels=3, cnn_num_filters=48,
                         num_classes_per_set=5, num_samples_per_class=1,
                         num_target_samples=15)
    else:
        overrides = dict(dataset_name="omniglot_dataset", image_height=28,
                         image_width=28, image_channels=1, cnn_num_filters=64,
                         num_classes_per_set=20, num_samples_per_class=5,
                         num_target_samples=1)
    args = get_args([])
    for k, v in overrides.items():
        setattr(args, k, v)
    args.batch_size = cli.tasks_per_gpu * world_size
    args.number_of_training_steps_per_iter = cli.inner_steps
    args.number_of_evaluation_steps_per_iter = cli.inner_steps
    args.second_order = cli.second_order in ("True", "true", True)
    args.first_order_to_second_order_epoch = -1
    args.use_multi_step_loss_optimization = True
    args.multi_step_loss_num_epochs = 15
    args.total_epochs = 100
    args.seed = 104
    args.synthetic_data = True
    return args


def main():
    cli = parse_cli()
    dist_ctx = init_distributed()
    world = dist_ctx.world_size
    args = build_args(cli, world)
    device = select_device(args)
    on_gpu = device.type == "cuda"

    model = MAMLFewShotClassifier(
        im_shape=(2, args.image_channels, args.image_height, args.image_width),
        device=device, args=args)
    model.attach_distributed(dist_ctx)

    stream = SyntheticEpisodeStream(args, rank=dist_ctx.rank, world_size=world)
    n_distinct = min(8, cli.steps + cli.warmup)
    batches = [tuple(t.to(device) for t in b)
               for b in stream.get_train_batches(n_distinct)]

    def step(i: int):
        model.run_train_iter(batches[i % n_distinct], epoch=0)

    for i in range(cli.warmup):
        step(i)

    dist_ctx.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(cli.steps):
        step(cli.warmup + i)
    if on_gpu:
        torch.cuda.synchronize()
    dist_ctx.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks (the slowest rank defines job time)
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64)
        if dist_ctx.backend == "nccl":
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tasks_per_iter = args.batch_size  # whole-job tasks per step
    value = tasks_per_iter * cli.steps / elapsed
    ms_per_step = elapsed / cli.steps * 1000.0

    if dist_ctx.rank == 0:
        print(json.dumps({
            "metric": "meta-tasks/sec",
            "value": value,
            "unit": "tasks/s",
            "n_gpus": world,
            "steps": cli.steps,
            "warmup": cli.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if on_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": cli.model,
                "global_batch": args.batch_size,
                "tasks_per_gpu": cli.tasks_per_gpu,
                "ways": args.num_classes_per_set,
                "shots": args.num_samples_per_class,
                "targets": args.num_target_samples,
                "image": [args.image_channels, args.image_height, args.image_width],
                "filters": args.cnn_num_filters,
                "inner_steps": cli.inner_steps,
                "second_order": bool(args.second_order),
                "msl": True,
                "parallelism": f"task-dp{world}",
            },
        }), flush=True)


if __name__ == "__main__":
    main()
