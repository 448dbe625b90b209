"""Statistical determinism hunt: run each candidate many times and count
bitwise mismatches vs the first result.  Targets the intermittent
conv0-weight-grad drift seen in det_bisect."""

import os
import sys

os.environ.setdefault("MAML355_DETERMINISTIC", "1")
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from howtotrainyourmamlpytorch_amd import ops  # noqa: E402
from howtotrainyourmamlpytorch_amd.config import get_args  # noqa: E402
from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream  # noqa: E402
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier  # noqa: E402

dev = torch.device("cuda", 0)


def hunt(tag, fn, trials):
    ref = fn()
    if isinstance(ref, torch.Tensor):
        ref = [ref]
    bad = 0
    for _ in range(trials):
        out = fn()
        if isinstance(out, torch.Tensor):
            out = [out]
        if any(not (a == b).all().item() for a, b in zip(ref, out)):
            bad += 1
    print(f"[{tag}] {bad}/{trials} mismatching trials", flush=True)


def main():
    ext = ops.hip_ext()
    torch.manual_seed(0)
    # engine shapes: conv0 support pass (C=1, tiny NB)
    T = 4
    x_sup = torch.randn(T, 5, 28, 28, 1, device=dev, dtype=torch.bfloat16)
    dy_sup = torch.randn(T, 5, 28, 28, 48, device=dev, dtype=torch.bfloat16)
    x_tgt = torch.randn(T, 15, 28, 28, 1, device=dev, dtype=torch.bfloat16)
    dy_tgt = torch.randn(T, 15, 28, 28, 48, device=dev, dtype=torch.bfloat16)
    w0 = torch.randn(T, 48, 1, 3, 3, device=dev)
    wp0 = ext.tconv_repack(w0, False)

    hunt("wgrad_C1_sup", lambda: [t.clone() for t in ext.tconv_wgrad(dy_sup, x_sup, 1, True)], 200)
    hunt("wgrad_C1_tgt", lambda: [t.clone() for t in ext.tconv_wgrad(dy_tgt, x_tgt, 1, True)], 200)
    hunt("mm_C1_sup", lambda: ext.tconv_mm(x_sup, wp0, None, 1, 28, 28, False)[0].clone(), 200)
    # conv1 shapes after pool (C=48, 14x14)
    x1 = torch.randn(T, 5, 14, 14, 48, device=dev, dtype=torch.bfloat16)
    dy1 = torch.randn(T, 5, 14, 14, 48, device=dev, dtype=torch.bfloat16)
    hunt("wgrad_C48_14", lambda: [t.clone() for t in ext.tconv_wgrad(dy1, x1, 1, True)], 100)

    gamma = (torch.rand(48, device=dev) + 0.5)
    beta = torch.randn(48, device=dev)
    x5 = torch.randn(T, 5, 28, 28, 48, device=dev, dtype=torch.bfloat16)
    y5, mask5, mean5, var5, rstd5 = ext.bn_act_pool_fwd(x5, gamma, beta, 1e-5, 0.01, None)
    dyp = torch.randn_like(y5)
    hunt("bnpool_fwd", lambda: [t.clone() for t in
                                ext.bn_act_pool_fwd(x5, gamma, beta, 1e-5, 0.01, None)[:1]], 100)
    hunt("bnpool_bwd", lambda: [t.clone() for t in
                                ext.bn_act_pool_bwd(dyp, mask5, x5, mean5, rstd5,
                                                    gamma, beta, 0.01)], 100)
    x3 = x5.reshape(T, -1, 48).contiguous()
    dy3 = torch.randn_like(x3)
    y_, mean_, var_, rstd_ = ext.bn_act_fwd(x3, gamma, beta, 1e-5, 0.01, True)
    hunt("bn_bwd_small", lambda: [t.clone() for t in
                                  ext.bn_act_bwd(dy3, x3, mean_, rstd_, gamma, beta, 0.01, True)], 100)
    gg = torch.randn(T, 48, device=dev)
    hunt("bn_dbwd_small", lambda: [t.clone() for t in ext.bn_act_dbwd(
        x3, dy3, dy3, mean_, rstd_, gamma, beta, gg, gg, 0.01, True)], 100)

    # linear at engine shapes (feat=48, bf16 bmm)
    from howtotrainyourmamlpytorch_amd.ops import reference as ref
    feats = torch.randn(T, 15, 48, device=dev, dtype=torch.bfloat16)
    lw = torch.randn(T, 5, 48, device=dev)
    lb = torch.randn(T, 5, device=dev)

    def lin():
        lw2 = lw.detach().clone().requires_grad_()
        f2 = feats.detach().clone().requires_grad_()
        out = ref.task_linear(f2, lw2.to(feats.dtype), lb.to(feats.dtype))
        out.float().square().sum().backward()
        return [out.detach().clone(), lw2.grad.clone(), f2.grad.clone()]
    hunt("linear48", lin, 100)

    # full engine meta-gradient, many trials
    args = get_args([
        "--batch_size", "4", "--num_classes_per_set", "5",
        "--num_samples_per_class", "1", "--num_target_samples", "3",
        "--image_height", "28", "--image_width", "28", "--image_channels", "1",
        "--cnn_num_filters", "48",
        "--number_of_training_steps_per_iter", "3",
        "--second_order", "True", "--first_order_to_second_order_epoch", "-1",
        "--total_epochs", "5", "--seed", "7", "--dataset_name", "synthetic",
    ])
    torch.manual_seed(123)
    model = MAMLFewShotClassifier(im_shape=(2, 1, 28, 28), device=dev, args=args)
    batch = next(iter(SyntheticEpisodeStream(args).get_train_batches(1)))

    def meta():
        losses, _ = model.train_forward_prop(batch, 0)
        gs = torch.autograd.grad(losses["loss"], model.trainable_parameters())
        torch.cuda.synchronize()
        return [g.clone() for g in gs]
    hunt("meta_grads_x30", meta, 30)


def hunt_trainloop_and_fresh_linear():
    """Targets the Adam-amplified full-train-loop nondeterminism: (a) the
    exact failing-test scenario repeated; (b) linear bmm with FRESH
    allocations per trial (allocator-address-dependent algorithm choice
    would escape same-tensor trials)."""
    from howtotrainyourmamlpytorch_amd.ops import reference as ref

    def train_once():
        torch.manual_seed(123)
        args = get_args([
            "--batch_size", "4", "--num_classes_per_set", "5",
            "--num_samples_per_class", "1", "--num_target_samples", "3",
            "--image_height", "28", "--image_width", "28", "--image_channels", "1",
            "--cnn_num_filters", "48",
            "--number_of_training_steps_per_iter", "3",
            "--second_order", "True", "--first_order_to_second_order_epoch", "-1",
            "--total_epochs", "5", "--seed", "7", "--dataset_name", "synthetic",
        ])
        model = MAMLFewShotClassifier(im_shape=(2, 1, 28, 28), device=dev, args=args)
        stream = SyntheticEpisodeStream(args)
        for batch in stream.get_train_batches(3):
            model.run_train_iter(batch, epoch=0)
        torch.cuda.synchronize()
        return model.classifier.theta.detach().clone()

    ref_theta = train_once()
    bad = 0
    for i in range(6):
        t = train_once()
        if not (t == ref_theta).all().item():
            bad += 1
            print(f"  trainloop trial {i}: max|d|={(t-ref_theta).abs().max().item():e}",
                  flush=True)
    print(f"[trainloop x6] {bad}/6 mismatching", flush=True)

    cpu_lw = torch.randn(4, 5, 48)
    cpu_f = torch.randn(4, 15, 48)
    ref_out = None
    bad = 0
    for i in range(50):
        # fresh GPU allocations with allocator perturbation
        junk = torch.randn((i % 7 + 1) * 333, device=dev)  # noqa: F841
        lw = cpu_lw.to(dev)
        f = cpu_f.to(dev).to(torch.bfloat16)
        lw2 = lw.requires_grad_()
        f2 = f.requires_grad_()
        out = ref.task_linear(f2, lw2.to(f.dtype), None)
        out.float().square().sum().backward()
        res = [out.detach().cpu(), lw2.grad.cpu(), f2.grad.cpu()]
        if ref_out is None:
            ref_out = res
        elif any(not (a == b).all().item() for a, b in zip(res, ref_out)):
            bad += 1
    print(f"[linear_freshalloc x50] {bad}/50 mismatching", flush=True)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "trainloop":
        hunt_trainloop_and_fresh_linear()
    else:
        main()
        hunt_trainloop_and_fresh_linear()
