"""Multi-process (gloo, world_size=2, CPU) tests of task-level data
parallelism: two ranks each training their task shard with the flat
all-reduce must produce exactly the single-process global-batch result."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from howtotrainyourmamlpytorch_amd.config import get_args


def dist_args():
    return get_args([
        "--batch_size", "4",
        "--num_classes_per_set", "3",
        "--num_samples_per_class", "1",
        "--num_target_samples", "2",
        "--image_height", "14", "--image_width", "14", "--image_channels", "1",
        "--cnn_num_filters", "4", "--num_stages", "3",
        "--number_of_training_steps_per_iter", "2",
        "--total_epochs", "2", "--total_iter_per_epoch", "2",
        "--seed", "3",
        "--synthetic_data", "True",
        "--dataset_name", "synthetic",
    ])


def _build_model(args, device):
    from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier
    torch.manual_seed(0)
    return MAMLFewShotClassifier(
        im_shape=(2, args.image_channels, args.image_height, args.image_width),
        device=device, args=args)


def _worker(rank, world, rdv_file, out_dir):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{rdv_file}",
                            rank=rank, world_size=world)
    from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream
    from howtotrainyourmamlpytorch_amd.parallel.dist import DistContext

    args = dist_args()
    device = torch.device("cpu")
    model = _build_model(args, device)
    ctx = DistContext(rank, world, rank, "gloo")
    model.attach_distributed(ctx)
    stream = SyntheticEpisodeStream(args, rank=rank, world_size=world)

    # iteration 0: capture the all-reduced meta-gradient (tight comparison;
    # Adam's early steps amplify fp noise to ~2*lr on near-zero elements)
    batch0 = next(iter(stream.get_train_batches(1)))
    losses, _ = model.train_forward_prop(batch0, epoch=0)
    model.optimizer.zero_grad()
    losses["loss"].backward()
    ctx.all_reduce_gradients(model.trainable_parameters())
    grad0 = torch.cat([p.grad.reshape(-1).clone()
                       for p in model.trainable_parameters()])
    model.optimizer.step()

    for batch in stream.get_train_batches(2):
        model.run_train_iter(batch, epoch=0)
    if rank == 0:
        torch.save({"theta": model.classifier.theta.detach(),
                    "lrs": model.inner_loop_lrs.detach(),
                    "grad0": grad0},
                   os.path.join(out_dir, "dist_result.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_two_rank_training_matches_single_process(tmp_path):
    rdv = str(tmp_path / "rdv")
    mp.spawn(_worker, args=(2, rdv, str(tmp_path)), nprocs=2, join=True)
    result = torch.load(tmp_path / "dist_result.pt", weights_only=False)

    # single-process run over the same global batch
    from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream
    args = dist_args()
    model = _build_model(args, torch.device("cpu"))
    stream = SyntheticEpisodeStream(args, rank=0, world_size=1)
    batch0 = next(iter(stream.get_train_batches(1)))
    losses, _ = model.train_forward_prop(batch0, epoch=0)
    model.optimizer.zero_grad()
    losses["loss"].backward()
    grad0 = torch.cat([p.grad.reshape(-1).clone()
                       for p in model.trainable_parameters()])
    model.optimizer.step()
    for batch in stream.get_train_batches(2):
        model.run_train_iter(batch, epoch=0)

    # the meta-gradient itself is exact up to fp reduction order
    torch.testing.assert_close(result["grad0"], grad0, rtol=1e-4, atol=1e-6)
    # params after 3 Adam steps: reduction-order noise near zero-grad
    # elements is amplified to ~2*lr by Adam's sign-like first steps
    torch.testing.assert_close(result["theta"], model.classifier.theta.detach(),
                               rtol=0.2, atol=7e-3)
    torch.testing.assert_close(result["lrs"], model.inner_loop_lrs.detach(),
                               rtol=0.2, atol=7e-3)


def _allreduce_worker(rank, world, rdv_file, out_dir):
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{rdv_file}",
                            rank=rank, world_size=world)
    from howtotrainyourmamlpytorch_amd.parallel.dist import DistContext
    ctx = DistContext(rank, world, rank, "gloo")
    v = ctx.all_reduce_scalar(float(rank + 1))
    assert abs(v - 1.5) < 1e-9
    p = torch.nn.Parameter(torch.ones(5))
    p.grad = torch.full((5,), float(rank))
    ctx.all_reduce_gradients([p])
    assert torch.allclose(p.grad, torch.full((5,), 0.5))
    dist.barrier()
    dist.destroy_process_group()


def test_allreduce_scalar_and_gradients(tmp_path):
    rdv = str(tmp_path / "rdv2")
    mp.spawn(_allreduce_worker, args=(2, rdv, str(tmp_path)), nprocs=2, join=True)
