"""RCCL (torch.distributed backend "nccl" on ROCm) smoke: two ranks
exercise the real collective code paths — flat-bucket all-reduce, the
(count, sum, sumsq) statistics reduction, and the overlapped per-chunk
reduction.  Measured on MI355X/RCCL 2.26: co-resident ranks on ONE device
are REFUSED ("Duplicate GPU detected"), so this needs >= 2 GPUs and runs
when the driver has a multi-GPU node; single-GPU boxes skip."""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _nccl_worker(rank, world, rdv_file, out_dir):
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    import torch.distributed as dist
    torch.cuda.set_device(rank)
    dist.init_process_group("nccl", init_method=f"file://{rdv_file}",
                            rank=rank, world_size=world)
    from howtotrainyourmamlpytorch_amd.parallel.dist import DistContext
    ctx = DistContext(rank, world, rank, "nccl")

    # flat-bucket all-reduce of parameter gradients
    torch.manual_seed(10 + rank)
    params = [torch.nn.Parameter(torch.randn(257, device="cuda")),
              torch.nn.Parameter(torch.randn(31, 5, device="cuda"))]
    grads_local = [torch.randn_like(p) for p in params]
    for p, g in zip(params, grads_local):
        p.grad = g.clone()
    ctx.all_reduce_gradients(params)

    # scalar + vector statistic reductions
    v = ctx.all_reduce_scalar(float(rank + 1))
    assert abs(v - (world + 1) / 2.0) < 1e-9
    n, s, q = ctx.all_reduce_sum_vector([1.0, float(rank), float(rank) ** 2])
    assert abs(n - world) < 1e-9

    # overlapped per-chunk reduction: two chunks, async handles
    ctx.start_overlapped_reduction(params)
    for p in params:
        p.grad = torch.ones_like(p) * (rank + 1)
    ctx.reduce_chunk_gradients(params)
    for p in params:
        p.grad = torch.ones_like(p) * 10.0
    ctx.reduce_chunk_gradients(params)
    ctx.finish_overlapped_reduction(params)
    # expected: mean over ranks of (rank+1) + 10 = 1.5 + 10 = 11.5
    for p in params:
        assert torch.allclose(p.grad, torch.full_like(p, 11.5)), p.grad.flatten()[:3]

    ctx.barrier()
    if rank == 0:
        torch.save({"ok": True}, os.path.join(out_dir, "rccl_ok.pt"))
    dist.destroy_process_group()


def test_rccl_two_ranks(tmp_path):
    if torch.cuda.device_count() < 2:
        pytest.skip("RCCL refuses co-resident ranks on one device "
                    "(verified on MI355X, RCCL 2.26); needs >= 2 GPUs")
    rdv = str(tmp_path / "rdv_nccl")
    ctx = mp.get_context("spawn")
    procs = []
    for r in range(2):
        p = ctx.Process(target=_nccl_worker, args=(r, 2, rdv, str(tmp_path)))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        assert p.exitcode == 0, f"rank failed with {p.exitcode}"
    assert (tmp_path / "rccl_ok.pt").is_file()


def test_rccl_single_rank_init_and_collectives(tmp_path):
    """1-rank RCCL process group on one MI355X: exercises RCCL comm
    creation and the DistContext collective code paths on hardware even
    when no second GPU exists (the 2-rank test above covers real
    communication when the driver has a multi-GPU node)."""
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", init_method=f"file://{tmp_path}/rdv1",
                            rank=0, world_size=1)
    try:
        from howtotrainyourmamlpytorch_amd.parallel.dist import DistContext
        ctx = DistContext(0, 1, 0, "nccl")
        p = torch.nn.Parameter(torch.randn(300, device="cuda"))
        p.grad = torch.randn_like(p)
        g0 = p.grad.clone()
        # world_size=1 short-circuits the flat bucket; call the raw
        # collective directly so RCCL actually executes
        t = p.grad.clone()
        dist.all_reduce(t)
        torch.testing.assert_close(t, g0)
        n, s, q = ctx.all_reduce_sum_vector([1.0, 2.0, 3.0])
        assert (n, s, q) == (1.0, 2.0, 3.0)
        ctx.start_overlapped_reduction([p])
        ctx.reduce_chunk_gradients([p])
        ctx.finish_overlapped_reduction([p])
        torch.testing.assert_close(p.grad, g0)
        ctx.barrier()
    finally:
        dist.destroy_process_group()
