"""Engine-level tests: shapes, a full train iter on CPU, and the central
correctness property of the MI355X design — task-batched execution is
numerically equivalent to serial per-task execution."""

import copy

import pytest
import torch

from howtotrainyourmamlpytorch_amd.config import get_args
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier


def tiny_args(**over):
    base = [
        "--batch_size", "2",
        "--num_classes_per_set", "3",
        "--num_samples_per_class", "2",
        "--num_target_samples", "2",
        "--image_height", "14", "--image_width", "14", "--image_channels", "1",
        "--cnn_num_filters", "8", "--num_stages", "3",
        "--number_of_training_steps_per_iter", "2",
        "--number_of_evaluation_steps_per_iter", "2",
        "--total_epochs", "4", "--total_iter_per_epoch", "3",
        "--multi_step_loss_num_epochs", "3",
        "--seed", "7",
    ]
    args = get_args(base)
    for k, v in over.items():
        setattr(args, k, v)
    return args


def make_batch(args, seed=0, tasks=None):
    g = torch.Generator().manual_seed(seed)
    B = tasks if tasks is not None else args.batch_size
    N, S, T = args.num_classes_per_set, args.num_samples_per_class, args.num_target_samples
    c, h, w = args.image_channels, args.image_height, args.image_width
    xs = torch.randn(B, N, S, c, h, w, generator=g)
    xt = torch.randn(B, N, T, c, h, w, generator=g)
    ys = torch.arange(N).view(1, N, 1).expand(B, N, S).contiguous()
    yt = torch.arange(N).view(1, N, 1).expand(B, N, T).contiguous()
    return xs, xt, ys, yt


def test_importance_vector_matches_reference_formula():
    args = tiny_args()
    args.number_of_training_steps_per_iter = 5
    args.multi_step_loss_num_epochs = 10
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14), device=torch.device("cpu"), args=args)
    model.current_epoch = 0
    v0 = model.get_per_step_loss_importance_vector()
    torch.testing.assert_close(v0, torch.full((5,), 0.2))
    model.current_epoch = 100  # far past annealing
    v = model.get_per_step_loss_importance_vector()
    torch.testing.assert_close(v[:4], torch.full((4,), 0.03 / 5))
    torch.testing.assert_close(v.sum(), torch.tensor(1.0), rtol=1e-5, atol=1e-6)


def test_train_iter_updates_params_and_returns_losses():
    args = tiny_args()
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14), device=torch.device("cpu"), args=args)
    batch = make_batch(args)
    theta_before = model.classifier.theta.detach().clone()
    lrs_before = model.inner_loop_lrs.detach().clone()
    losses, preds = model.run_train_iter(batch, epoch=0)
    assert "loss" in losses and "accuracy" in losses and "learning_rate" in losses
    assert preds.shape == (2, 3 * 2, 3)
    assert not torch.allclose(model.classifier.theta.detach(), theta_before)
    # LSLR learning rates are meta-learned too
    assert not torch.allclose(model.inner_loop_lrs.detach(), lrs_before)
    # second iter at later epoch changes the outer LR via cosine annealing
    losses2, _ = model.run_train_iter(batch, epoch=2)
    assert losses2["learning_rate"] < losses["learning_rate"]


def test_validation_iter_restores_bn_stats():
    args = tiny_args()
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14), device=torch.device("cpu"), args=args)
    batch = make_batch(args)
    model.run_train_iter(batch, epoch=0)
    rm = model.classifier.bn_running_mean_0.detach().clone()
    model.run_validation_iter(make_batch(args, seed=3))
    torch.testing.assert_close(model.classifier.bn_running_mean_0.detach(), rm)


def test_first_order_vs_second_order_differ():
    args = tiny_args()
    batch = make_batch(args)
    results = {}
    for fo_epoch in (-1, 100):  # -1: second-order from start; 100: first-order
        a = tiny_args()
        a.first_order_to_second_order_epoch = fo_epoch
        m = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14), device=torch.device("cpu"), args=a)
        losses, _ = m.run_train_iter(batch, epoch=1)
        results[fo_epoch] = m.classifier.theta.detach().clone()
    assert not torch.allclose(results[-1], results[100], rtol=1e-5, atol=1e-7)


@pytest.mark.parametrize("second_order", [False, True])
def test_task_batched_equals_serial(second_order):
    """THE core equivalence: running B tasks batched must produce the same
    outer loss and the same meta-gradient as running them one at a time and
    averaging (what the reference's serial loop computes)."""
    args = tiny_args()
    args.second_order = second_order
    args.batch_size = 3
    batch = make_batch(args, tasks=3)

    def fresh():
        torch.manual_seed(0)
        return MAMLFewShotClassifier(im_shape=(2, 1, 14, 14),
                                     device=torch.device("cpu"), args=args)

    # batched
    mb = fresh()
    losses_b, _ = mb.train_forward_prop(batch, epoch=1)
    gb = torch.autograd.grad(losses_b["loss"], mb.classifier.theta, retain_graph=False)[0]

    # serial: same init (fresh() is seeded), one task at a time
    ms = fresh()
    torch.testing.assert_close(mb.classifier.theta, ms.classifier.theta)
    per_task_losses = []
    grads = torch.zeros_like(ms.classifier.theta)
    for t in range(3):
        sub = tuple(x[t:t + 1] for x in batch)
        losses_s, _ = ms.train_forward_prop(sub, epoch=1)
        per_task_losses.append(losses_s["loss"].detach())
        g = torch.autograd.grad(losses_s["loss"], ms.classifier.theta)[0]
        grads += g / 3.0
    serial_loss = torch.stack(per_task_losses).mean()

    torch.testing.assert_close(losses_b["loss"].detach(), serial_loss, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(gb, grads, rtol=1e-4, atol=1e-6)


def test_msl_weighting_active_then_inactive():
    args = tiny_args()
    model = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14), device=torch.device("cpu"), args=args)
    batch = make_batch(args)
    losses_active, _ = model.run_train_iter(batch, epoch=0)
    assert "loss_importance_vector_0" in losses_active
    losses_late, _ = model.run_train_iter(batch, epoch=3)  # >= multi_step_loss_num_epochs
    assert "loss_importance_vector_0" not in losses_late


def test_chunked_outer_backward_matches_unchunked():
    """task_chunk_size accumulation must reproduce the single-graph meta
    update exactly (outer loss is a mean over tasks)."""
    args = tiny_args()
    args.batch_size = 4
    batch = make_batch(args, tasks=4)

    def run(chunk):
        torch.manual_seed(0)
        a = tiny_args()
        a.batch_size = 4
        a.task_chunk_size = chunk
        m = MAMLFewShotClassifier(im_shape=(2, 1, 14, 14),
                                  device=torch.device("cpu"), args=a)
        m.optimizer.step = lambda: None  # compare pre-Adam gradients
        losses, _ = m.run_train_iter(batch, epoch=1)
        grads = torch.cat([p.grad.reshape(-1) for p in m.trainable_parameters()])
        return losses, grads

    losses_full, g_full = run(0)
    losses_chunk, g_chunk = run(2)
    assert abs(losses_full["loss"] - losses_chunk["loss"]) < 1e-6
    # accumulation order differs between chunked and single-graph backward;
    # gradients agree to fp32 noise (Adam's early steps would amplify that
    # noise to ~2*lr on near-zero elements, so params are not compared)
    torch.testing.assert_close(g_chunk, g_full, rtol=1e-4, atol=1e-6)


def test_fused_adam_matches_torch_adam_cpu():
    """FusedAdam (CPU path = ops.reference.fused_adam_step) matches
    torch.optim.Adam over several steps, and its state dict round-trips
    through a plain torch.optim.Adam."""
    import copy
    torch.manual_seed(0)
    from howtotrainyourmamlpytorch_amd.meta.fused_adam import FusedAdam
    p1 = [torch.nn.Parameter(torch.randn(7, 3)), torch.nn.Parameter(torch.randn(11))]
    p2 = [torch.nn.Parameter(p.detach().clone()) for p in p1]
    opt1 = FusedAdam(p1, lr=0.01)
    opt2 = torch.optim.Adam(p2, lr=0.01, amsgrad=False)
    for it in range(5):
        g = [torch.randn_like(p) for p in p1]
        for p, gg in zip(p1, g):
            p.grad = gg.clone()
        for p, gg in zip(p2, g):
            p.grad = gg.clone()
        opt1.step()
        opt2.step()
    for a, b in zip(p1, p2):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)
    # state-dict interchange: torch Adam accepts FusedAdam's state
    opt3 = torch.optim.Adam([torch.nn.Parameter(p.detach().clone()) for p in p1],
                            lr=0.01)
    opt3.load_state_dict(copy.deepcopy(opt1.state_dict()))


def test_fused_adam_clamp_matches_manual_clamp():
    """grad_clamp in FusedAdam == reference's p.grad.clamp_(-c,c) + Adam."""
    torch.manual_seed(1)
    from howtotrainyourmamlpytorch_amd.meta.fused_adam import FusedAdam
    p1 = [torch.nn.Parameter(torch.randn(13))]
    p2 = [torch.nn.Parameter(p1[0].detach().clone())]
    opt1 = FusedAdam(p1, lr=0.05, grad_clamp=0.5)
    opt2 = torch.optim.Adam(p2, lr=0.05)
    for it in range(3):
        g = torch.randn(13) * 2.0
        p1[0].grad = g.clone()
        p2[0].grad = g.clamp(-0.5, 0.5)
        opt1.step()
        opt2.step()
    torch.testing.assert_close(p1[0], p2[0], rtol=1e-5, atol=1e-6)


def test_fused_adam_mixed_step_counts():
    """Params with differing step counts (after a partial state load) are
    bucketed and updated per their own bias correction."""
    torch.manual_seed(3)
    from howtotrainyourmamlpytorch_amd.meta.fused_adam import FusedAdam
    p1 = torch.nn.Parameter(torch.randn(5))
    p2 = torch.nn.Parameter(torch.randn(5))
    opt = FusedAdam([p1, p2], lr=0.1)
    p1.grad = torch.ones(5)
    p2.grad = torch.ones(5)
    opt.step()
    # simulate a partial load: p2's state restarts
    opt.state[p2]["step"] = torch.zeros(())
    opt.state[p2]["exp_avg"].zero_()
    opt.state[p2]["exp_avg_sq"].zero_()
    p1.grad = torch.ones(5)
    p2.grad = torch.ones(5)
    opt.step()  # p1 at step 2, p2 at step 1 -> two buckets, no crash
    ref = torch.nn.Parameter(torch.zeros(5))
    assert opt.state[p1]["step"].item() == 2.0
    assert opt.state[p2]["step"].item() == 1.0
