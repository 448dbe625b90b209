"""Determinism bisect (run on MI355X): runs each op and the full
meta-gradient twice with identical inputs and reports which results are
not bitwise-identical.  Localizes residual run-to-run nondeterminism."""

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


def check(tag, fn):
    a = fn()
    b = fn()
    if isinstance(a, torch.Tensor):
        a, b = [a], [b]
    bad = []
    for i, (x, y) in enumerate(zip(a, b)):
        if not (x == y).all().item():
            bad.append((i, (x - y).abs().max().item()))
    print(f"[{tag}] {'OK' if not bad else 'DIFF ' + str(bad)}", flush=True)


def main():
    ext = ops.hip_ext()
    torch.manual_seed(0)
    T, NB, H, W, C, F = 4, 15, 28, 28, 48, 48
    x = torch.randn(T, NB, H, W, C, device=dev, dtype=torch.bfloat16)
    w = torch.randn(T, F, C, 3, 3, device=dev)
    bsz = torch.randn(T, F, device=dev)
    wp = ext.tconv_repack(w, False)
    dy = torch.randn(T, NB, H, W, F, device=dev, dtype=torch.bfloat16)

    check("conv_fwd", lambda: ext.tconv_mm(x, wp, bsz, 1, H, W, False)[0].clone())
    check("wgrad_det", lambda: [t.clone() for t in ext.tconv_wgrad(dy, x, 1, True)])
    gamma = (torch.rand(F, device=dev) + 0.5)
    beta = torch.randn(F, device=dev)
    x3 = x.reshape(T, -1, C).contiguous()
    check("bn_fwd", lambda: [t.clone() for t in ext.bn_act_fwd(x3, gamma, beta, 1e-5, 0.01, True)])
    y, mean, var, rstd = ext.bn_act_fwd(x3, gamma, beta, 1e-5, 0.01, True)
    dy3 = dy.reshape(T, -1, F).contiguous()
    check("bn_bwd", lambda: [t.clone() for t in ext.bn_act_bwd(dy3, x3, mean, rstd, gamma, beta, 0.01, True)])
    gg = torch.randn(T, F, device=dev)
    check("bn_dbwd", lambda: [t.clone() for t in ext.bn_act_dbwd(
        x3, dy3, dy3, mean, rstd, gamma, beta, gg, gg, 0.01, True)])
    logits = torch.randn(T, 75, 5, device=dev)
    labels = torch.randint(0, 5, (T, 75), device=dev)
    check("ce_fwd", lambda: [t.clone() for t in ext.ce_fwd(logits, labels)])
    feats = torch.randn(T, 75, 1200, device=dev, dtype=torch.bfloat16)
    lw = torch.randn(T, 5, 1200, device=dev)
    lb = torch.randn(T, 5, device=dev)
    from howtotrainyourmamlpytorch_amd.ops import reference as ref
    check("linear_fwd", lambda: ref.task_linear(
        feats, lw.to(feats.dtype), lb.to(feats.dtype)).clone())

    def linear_grads():
        lw2 = lw.detach().clone().requires_grad_()
        f2 = feats.detach().clone().requires_grad_()
        out = ref.task_linear(f2, lw2.to(feats.dtype), lb.to(feats.dtype))
        out.float().square().sum().backward()
        return [lw2.grad.clone(), f2.grad.clone()]
    check("linear_bwd", linear_grads)

    # full meta-gradient
    def make_args(channels=1, second="True", steps=3, msl="True"):
        return get_args([
            "--batch_size", "4", "--num_classes_per_set", "5",
            "--num_samples_per_class", "1", "--num_target_samples", "3",
            "--image_height", "28", "--image_width", "28",
            "--image_channels", str(channels),
            "--cnn_num_filters", "48",
            "--number_of_training_steps_per_iter", str(steps),
            "--use_multi_step_loss_optimization", msl,
            "--second_order", second, "--first_order_to_second_order_epoch", "-1",
            "--total_epochs", "5", "--seed", "7", "--dataset_name", "synthetic",
        ])
    args = make_args()

    def meta_grads_for(a):
        torch.manual_seed(123)
        model = MAMLFewShotClassifier(
            im_shape=(2, a.image_channels, 28, 28), device=dev, args=a)
        batch = next(iter(SyntheticEpisodeStream(a).get_train_batches(1)))
        losses, _ = model.train_forward_prop(batch, 0)
        gs = torch.autograd.grad(losses["loss"], model.trainable_parameters())
        torch.cuda.synchronize()
        return [g.clone() for g in gs]

    for tag, a in [("v1_firstorder", make_args(second="False")),
                   ("v2_c8", make_args(channels=8)),
                   ("v3_steps1", make_args(steps=1)),
                   ("v4_nomsl", make_args(msl="False"))]:
        ga = meta_grads_for(a)
        gb = meta_grads_for(a)
        bad = [i for i, (x_, y_) in enumerate(zip(ga, gb))
               if not (x_ == y_).all().item()]
        print(f"[meta_variant {tag}] {'OK' if not bad else 'DIFF at ' + str(bad)}",
              flush=True)

    def meta_grads():
        return meta_grads_for(args)
    # first-layer (C=1) wgrad determinism at engine shapes
    x1 = torch.randn(4, 15, 28, 28, 1, device=dev, dtype=torch.bfloat16)
    dy1 = torch.randn(4, 15, 28, 28, 48, device=dev, dtype=torch.bfloat16)
    check("wgrad_C1_det", lambda: [t.clone() for t in ext.tconv_wgrad(dy1, x1, 1, True)])
    probs = torch.softmax(logits, -1)
    gt = torch.randn(T, device=dev)
    check("ce_bwd", lambda: ext.ce_bwd(probs, labels, gt).clone())
    gdl = torch.randn_like(probs)
    check("ce_dbwd", lambda: [t.clone() for t in ext.ce_dbwd(probs, labels, gdl, gt)])
    x4 = torch.randn(60, 28, 28, 48, device=dev, dtype=torch.bfloat16)
    yp, maskp = ext.maxpool2x2_fwd(x4)
    dyp = torch.randn_like(yp)
    check("pool", lambda: [ext.maxpool2x2_fwd(x4)[0].clone(),
                           ext.maxpool2x2_bwd(dyp, maskp, 28, 28).clone()])

    names = None
    a = meta_grads()
    b = meta_grads()
    model = MAMLFewShotClassifier(im_shape=(2, 1, 28, 28), device=dev, args=args)
    names = [n for n, p in model.named_parameters() if p.requires_grad]
    for n, x_, y_ in zip(names, a, b):
        eq = (x_ == y_).all().item()
        print(f"[meta_grad {n}] {'OK' if eq else 'DIFF %e' % (x_ - y_).abs().max().item()}",
              flush=True)
        if not eq and n == "classifier.theta":
            for s in model.classifier.arena.specs:
                d = (x_[s.offset:s.offset + s.numel]
                     - y_[s.offset:s.offset + s.numel]).abs().max().item()
                print(f"    theta[{s.name}] {'OK' if d == 0 else 'DIFF %e' % d}",
                      flush=True)


if __name__ == "__main__":
    main()
