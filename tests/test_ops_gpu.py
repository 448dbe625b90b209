"""GPU numerics: every HIP kernel against the pure-torch fp32 reference op,
including backward and (where the engine needs it) double-backward paths.
All tests require a real MI355X (run via gpurun)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from howtotrainyourmamlpytorch_amd import ops
from howtotrainyourmamlpytorch_amd.ops import reference as ref


def dev():
    return torch.device("cuda", 0)


def test_extension_is_loaded_and_mandatory():
    assert ops.hip_ext() is not None, "HIP extension must load on a GPU box"


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-5), (torch.bfloat16, 2e-2)])
@pytest.mark.parametrize("per_task", [False, True])
def test_bn_act_fwd_matches_reference(dtype, tol, per_task):
    torch.manual_seed(0)
    T, NS, H, W, C = 3, 7, 9, 9, 48
    x = torch.randn(T, NS, H, W, C, device=dev(), dtype=dtype)
    gamma = (torch.rand(T, C) + 0.5) if per_task else (torch.rand(C) + 0.5)
    beta = torch.randn(T, C) if per_task else torch.randn(C)
    gamma, beta = gamma.to(dev()), beta.to(dev())
    y, mean, var = ops.task_bn_act(x, gamma, beta)
    yr, mr, vr = ref.task_bn_act(x.float().cpu(), gamma.float().cpu(), beta.float().cpu())
    torch.testing.assert_close(y.float().cpu(), yr, rtol=tol, atol=tol)
    torch.testing.assert_close(mean.cpu(), mr, rtol=tol, atol=tol)
    torch.testing.assert_close(var.cpu(), vr, rtol=tol, atol=tol)


def test_bn_act_backward_matches_reference():
    torch.manual_seed(1)
    T, NS, H, W, C = 2, 5, 6, 6, 64
    x = torch.randn(T, NS, H, W, C, device=dev(), requires_grad=True)
    gamma = (torch.rand(C, device=dev()) + 0.5).requires_grad_(True)
    beta = torch.randn(C, device=dev()).requires_grad_(True)
    y, _, _ = ops.task_bn_act(x, gamma, beta)
    g = torch.randn_like(y)
    dx, dgamma, dbeta = torch.autograd.grad(y, (x, gamma, beta), g)

    xr = x.detach().float().cpu().requires_grad_(True)
    gr = gamma.detach().float().cpu().requires_grad_(True)
    br = beta.detach().float().cpu().requires_grad_(True)
    yr, _, _ = ref.task_bn_act(xr, gr, br)
    dxr, dgr, dbr = torch.autograd.grad(yr, (xr, gr, br), g.float().cpu())
    torch.testing.assert_close(dx.cpu(), dxr, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dgamma.cpu(), dgr, rtol=1e-4, atol=1e-3)
    torch.testing.assert_close(dbeta.cpu(), dbr, rtol=1e-4, atol=1e-3)


def test_bn_act_double_backward_matches_reference():
    torch.manual_seed(2)
    T, NS, H, W, C = 2, 3, 4, 4, 48
    x = torch.randn(T, NS, H, W, C, device=dev(), requires_grad=True)
    gamma = (torch.rand(C, device=dev()) + 0.5).requires_grad_(True)
    beta = torch.randn(C, device=dev()).requires_grad_(True)

    def loss_of_grad(op, x_, g_, b_):
        y, _, _ = op(x_, g_, b_)
        l = (y ** 2).mean()
        (gx,) = torch.autograd.grad(l, (x_,), create_graph=True)
        return (gx ** 2).sum()

    l2 = loss_of_grad(ops.task_bn_act, x, gamma, beta)
    gg = torch.autograd.grad(l2, (x, gamma, beta))

    xr = x.detach().float().cpu().requires_grad_(True)
    gr = gamma.detach().float().cpu().requires_grad_(True)
    br = beta.detach().float().cpu().requires_grad_(True)
    l2r = loss_of_grad(ref.task_bn_act, xr, gr, br)
    ggr = torch.autograd.grad(l2r, (xr, gr, br))
    for a, b in zip(gg, ggr):
        torch.testing.assert_close(a.cpu(), b, rtol=1e-3, atol=1e-3)


@pytest.mark.parametrize("H,W", [(8, 8), (7, 9)])
def test_maxpool_fwd_bwd_double(H, W):
    torch.manual_seed(3)
    T, NS, C = 2, 4, 48
    x = torch.randn(T, NS, H, W, C, device=dev(), requires_grad=True)
    y = ops.task_maxpool2x2(x)
    yr = ref.task_maxpool2x2(x.detach().cpu())
    torch.testing.assert_close(y.detach().cpu(), yr)
    # double-backward flows through grad_output only (the argmax mask is a
    # discrete function of x), so g must require grad to exercise it
    g = torch.randn_like(y).requires_grad_(True)
    (dx,) = torch.autograd.grad(y, x, g, create_graph=True)
    xr = x.detach().cpu().requires_grad_(True)
    gr = g.detach().cpu().requires_grad_(True)
    yr2 = ref.task_maxpool2x2(xr)
    (dxr,) = torch.autograd.grad(yr2, xr, gr, create_graph=True)
    torch.testing.assert_close(dx.detach().cpu(), dxr.detach())
    # double backward: d/dg of ||dx||^2  (gather path of _PoolBwdFn)
    (ddg,) = torch.autograd.grad((dx ** 2).sum(), g)
    (ddgr,) = torch.autograd.grad((dxr ** 2).sum(), gr)
    torch.testing.assert_close(ddg.cpu(), ddgr, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("ways", [5, 20])
def test_ce_fwd_bwd_and_create_graph(ways):
    torch.manual_seed(4)
    T, M = 3, 40
    logits = torch.randn(T, M, ways, device=dev(), requires_grad=True)
    labels = torch.randint(0, ways, (T, M), device=dev())
    loss = ops.softmax_cross_entropy(logits, labels)
    lossr = ref.softmax_cross_entropy(logits.detach().cpu(), labels.cpu())
    torch.testing.assert_close(loss.detach().cpu(), lossr, rtol=1e-5, atol=1e-5)
    gt = torch.rand(T, device=dev())
    (dl,) = torch.autograd.grad((loss * gt).sum(), logits, retain_graph=True)
    lr = logits.detach().cpu().requires_grad_(True)
    (dlr,) = torch.autograd.grad((ref.softmax_cross_entropy(lr, labels.cpu())
                                  * gt.cpu()).sum(), lr)
    torch.testing.assert_close(dl.cpu(), dlr, rtol=1e-5, atol=1e-5)
    # create_graph: grad-of-grad of CE wrt logits
    (dl2,) = torch.autograd.grad(loss.sum(), logits, create_graph=True)
    (ddl,) = torch.autograd.grad((dl2 ** 2).sum(), logits)
    lr2 = logits.detach().cpu().requires_grad_(True)
    (dlr2,) = torch.autograd.grad(ref.softmax_cross_entropy(lr2, labels.cpu()).sum(),
                                  lr2, create_graph=True)
    (ddlr,) = torch.autograd.grad((dlr2 ** 2).sum(), lr2)
    torch.testing.assert_close(ddl.cpu(), ddlr, rtol=1e-4, atol=1e-5)


def test_lslr_update_fwd_bwd():
    torch.manual_seed(5)
    T, P = 4, 1000
    arena = torch.randn(T, P, device=dev(), requires_grad=True)
    grad = torch.randn(T, P, device=dev(), requires_grad=True)
    lr = torch.rand(P, device=dev(), requires_grad=True)
    out = ops.lslr_update(arena, grad, lr)
    torch.testing.assert_close(out.detach().cpu(),
                               ref.lslr_update(arena.detach().cpu(),
                                               grad.detach().cpu(), lr.detach().cpu()))
    g = torch.randn_like(out)
    da, dg, dl = torch.autograd.grad(out, (arena, grad, lr), g)
    torch.testing.assert_close(da.cpu(), g.cpu())
    torch.testing.assert_close(dg.cpu(), (-lr.detach().unsqueeze(0) * g).cpu())
    torch.testing.assert_close(dl.cpu(), (-(g * grad.detach()).sum(0)).cpu(),
                               rtol=1e-4, atol=1e-4)


def _engine_build(device, compute_dtype="bf16", fp32_support="False"):
    from howtotrainyourmamlpytorch_amd.config import get_args
    from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier

    args = get_args([
        "--batch_size", "2", "--num_classes_per_set", "3",
        "--num_samples_per_class", "2", "--num_target_samples", "2",
        "--image_height", "14", "--image_width", "14", "--image_channels", "1",
        "--cnn_num_filters", "8", "--num_stages", "3",
        "--number_of_training_steps_per_iter", "2",
        "--seed", "11", "--compute_dtype", compute_dtype,
        "--fp32_support_pass", fp32_support,
    ])
    torch.manual_seed(0)
    return args, MAMLFewShotClassifier(im_shape=(2, 1, 14, 14),
                                       device=device, args=args)


def _engine_batch():
    g = torch.Generator().manual_seed(9)
    xs = torch.randn(2, 3, 2, 1, 14, 14, generator=g)
    xt = torch.randn(2, 3, 2, 1, 14, 14, generator=g)
    ys = torch.arange(3).view(1, 3, 1).expand(2, 3, 2).contiguous()
    yt = ys.clone()
    return (xs, xt, ys, yt)


def _theta_grad(model, batch):
    losses, _ = model.train_forward_prop(batch, epoch=1)
    g = torch.autograd.grad(losses["loss"], model.classifier.theta)[0]
    return float(losses["loss"].detach()), g


def test_engine_gpu_fp32_matches_cpu_tight():
    """GPU fp32 eager path (kernels disabled) vs CPU fp32 oracle — any
    difference beyond reduction-order noise is a real bug."""
    from howtotrainyourmamlpytorch_amd import ops as opsmod
    batch = _engine_batch()
    _, m_cpu = _engine_build(torch.device("cpu"))
    loss_c, gc = _theta_grad(m_cpu, batch)
    opsmod.disable_hip_kernels()
    try:
        _, m_gpu = _engine_build(dev(), compute_dtype="fp32")
        loss_g, gg = _theta_grad(m_gpu, batch)
    finally:
        opsmod.enable_hip_kernels()
    assert abs(loss_c - loss_g) < 1e-4
    torch.testing.assert_close(gg.cpu(), gc, rtol=1e-3, atol=1e-5)


def test_engine_gpu_bf16_kernels_close_to_cpu():
    """GPU bf16 HIP-kernel path vs CPU fp32 oracle: second-order
    meta-gradient direction must survive bf16 (relative L2 + cosine)."""
    batch = _engine_batch()
    _, m_cpu = _engine_build(torch.device("cpu"))
    loss_c, gc = _theta_grad(m_cpu, batch)
    _, m_gpu = _engine_build(dev())
    loss_g, gg = _theta_grad(m_gpu, batch)
    assert abs(loss_c - loss_g) < 3e-2
    gg = gg.cpu()
    rel_l2 = (gg - gc).norm() / gc.norm()
    cos = torch.nn.functional.cosine_similarity(gg.flatten(), gc.flatten(), dim=0)
    # measured ~0.22 rel-L2 for a 2-step second-order graph in bf16 — the
    # gradient *direction* is what training needs (learning test covers
    # end-to-end); fp32 tight equivalence is asserted separately above
    assert rel_l2 < 0.40, f"relative L2 {rel_l2:.4f}"
    assert cos > 0.95, f"cosine {cos:.5f}"


def test_bn_act_pool_fused_matches_composition():
    torch.manual_seed(7)
    T, NS, H, W, C = 3, 5, 10, 10, 48
    x = torch.randn(T, NS, H, W, C, device=dev(), requires_grad=True)
    gamma = (torch.rand(C, device=dev()) + 0.5).requires_grad_(True)
    beta = torch.randn(C, device=dev()).requires_grad_(True)

    y_f, mean_f, var_f = ops.task_bn_act_pool(x, gamma, beta)
    y_c0, mean_c, var_c = ops.task_bn_act(x, gamma, beta)
    y_c = ops.task_maxpool2x2(y_c0)
    torch.testing.assert_close(y_f, y_c, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(mean_f, mean_c, rtol=1e-5, atol=1e-6)

    g = torch.randn_like(y_f)
    gf = torch.autograd.grad(y_f, (x, gamma, beta), g, retain_graph=True)
    gc = torch.autograd.grad(y_c, (x, gamma, beta), g, retain_graph=True)
    for a, b in zip(gf, gc):
        torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-4)

    # second-order through the fused op matches the composition
    def grad_norm(y_, x_):
        (gx,) = torch.autograd.grad((y_.float() ** 2).mean(), (x_,), create_graph=True)
        return (gx.float() ** 2).sum()

    x2 = x.detach().requires_grad_(True)
    y2, _, _ = ops.task_bn_act_pool(x2, gamma.detach(), beta.detach())
    (gg_f,) = torch.autograd.grad(grad_norm(y2, x2), (x2,))
    x3 = x.detach().requires_grad_(True)
    y3 = ops.task_maxpool2x2(ops.task_bn_act(x3, gamma.detach(), beta.detach())[0])
    (gg_c,) = torch.autograd.grad(grad_norm(y3, x3), (x3,))
    torch.testing.assert_close(gg_f, gg_c, rtol=1e-3, atol=1e-4)


def test_bn_act_pool_fused_per_task_affine():
    """inner-loop-optimizable BN params: per-task gamma/beta through the
    fused BN+act+pool path."""
    torch.manual_seed(8)
    T, NS, H, W, C = 2, 3, 8, 8, 48
    x = torch.randn(T, NS, H, W, C, device=dev(), requires_grad=True)
    gamma = (torch.rand(T, C, device=dev()) + 0.5).requires_grad_(True)
    beta = torch.randn(T, C, device=dev()).requires_grad_(True)
    y_f, _, _ = ops.task_bn_act_pool(x, gamma, beta)
    y_c = ops.task_maxpool2x2(ops.task_bn_act(x, gamma, beta)[0])
    torch.testing.assert_close(y_f, y_c, rtol=1e-4, atol=1e-4)
    g = torch.randn_like(y_f)
    gf = torch.autograd.grad(y_f, (x, gamma, beta), g, retain_graph=True)
    gc = torch.autograd.grad(y_c, (x, gamma, beta), g)
    for a, b in zip(gf, gc):
        torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-4)


def test_bn_act_double_backward_per_task_affine():
    """Second-order through BN with inner-loop-adapted (per-task) gamma/beta
    — the PER_TASK_AFFINE template path of bn_dbwd."""
    torch.manual_seed(9)
    T, NS, H, W, C = 2, 3, 4, 4, 48
    x = torch.randn(T, NS, H, W, C, device=dev(), requires_grad=True)
    gamma = (torch.rand(T, C, device=dev()) + 0.5).requires_grad_(True)
    beta = torch.randn(T, C, device=dev()).requires_grad_(True)

    def loss_of_grad(op, x_, g_, b_):
        y, _, _ = op(x_, g_, b_)
        l = (y.float() ** 2).mean()
        gx, gg = torch.autograd.grad(l, (x_, g_), create_graph=True)
        return (gx.float() ** 2).sum() + (gg.float() ** 2).sum()

    l2 = loss_of_grad(ops.task_bn_act, x, gamma, beta)
    gg = torch.autograd.grad(l2, (x, gamma))

    xr = x.detach().float().cpu().requires_grad_(True)
    gr = gamma.detach().float().cpu().requires_grad_(True)
    br = beta.detach().float().cpu().requires_grad_(True)
    l2r = loss_of_grad(ref.task_bn_act, xr, gr, br)
    ggr = torch.autograd.grad(l2r, (xr, gr))
    for a, b in zip(gg, ggr):
        torch.testing.assert_close(a.cpu(), b, rtol=2e-3, atol=2e-3)


def test_fused_adam_kernel_matches_reference():
    """adam.hip multi-tensor kernel vs ops.reference.fused_adam_step on
    identical fp32 state, including the fused grad clamp."""
    torch.manual_seed(5)
    shapes = [(1000,), (64, 9), (7,), (3, 5, 5)]
    pg = [torch.randn(*s) for s in shapes]
    gg = [torch.randn(*s) * 3.0 for s in shapes]
    mm = [torch.randn(*s).abs() * 0.1 for s in shapes]
    vv = [torch.rand(*s) * 0.01 for s in shapes]

    p_c = [t.clone() for t in pg]
    m_c = [t.clone() for t in mm]
    v_c = [t.clone() for t in vv]
    ref.fused_adam_step(p_c, [t.clone() for t in gg], m_c, v_c,
                        step=3, lr=0.01, clamp=1.5)

    p_g = [t.clone().to(dev()) for t in pg]
    g_g = [t.clone().to(dev()) for t in gg]
    m_g = [t.clone().to(dev()) for t in mm]
    v_g = [t.clone().to(dev()) for t in vv]
    ops.fused_adam_step(p_g, g_g, m_g, v_g, step=3, lr=0.01, clamp=1.5)
    for a, b in zip(p_g, p_c):
        torch.testing.assert_close(a.cpu(), b, rtol=1e-5, atol=1e-6)
    for a, b in zip(m_g, m_c):
        torch.testing.assert_close(a.cpu(), b, rtol=1e-5, atol=1e-6)
    for a, b in zip(v_g, v_c):
        torch.testing.assert_close(a.cpu(), b, rtol=1e-5, atol=1e-6)


def test_fused_adam_optimizer_runs_on_gpu():
    """FusedAdam drives the HIP kernel end-to-end on CUDA params and keeps
    torch state-dict format."""
    from howtotrainyourmamlpytorch_amd.meta.fused_adam import FusedAdam
    torch.manual_seed(6)
    p_gpu = [torch.nn.Parameter(torch.randn(50, device=dev())),
             torch.nn.Parameter(torch.randn(8, 8, device=dev()))]
    p_cpu = [torch.nn.Parameter(p.detach().cpu().clone()) for p in p_gpu]
    opt_g = FusedAdam(p_gpu, lr=0.02)
    opt_c = torch.optim.Adam(p_cpu, lr=0.02)
    for it in range(4):
        g = [torch.randn_like(p) for p in p_cpu]
        for p, gg_ in zip(p_gpu, g):
            p.grad = gg_.to(dev())
        for p, gg_ in zip(p_cpu, g):
            p.grad = gg_.clone()
        opt_g.step()
        opt_c.step()
    for a, b in zip(p_gpu, p_cpu):
        torch.testing.assert_close(a.detach().cpu(), b.detach(), rtol=1e-4, atol=1e-5)
    sd = opt_g.state_dict()
    assert "exp_avg" in list(sd["state"].values())[0]


def test_linear_trio_matches_reference():
    """lin_fwd/dx/wgrad vs the torch oracle incl. second-order compose."""
    torch.manual_seed(21)
    T, M, K, ways = 3, 17, 200, 5
    x = torch.randn(T, M, K, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(T, ways, K, device=dev(), requires_grad=True)
    b = torch.randn(T, ways, device=dev(), requires_grad=True)
    y = ops.task_linear(x, w, b)
    xr = x.detach().float().cpu().requires_grad_()
    wr = w.detach().cpu().requires_grad_()
    br = b.detach().cpu().requires_grad_()
    yr = ref.task_linear(xr, wr.to(torch.bfloat16).float(), br.to(torch.bfloat16).float())
    torch.testing.assert_close(y.float().cpu(), yr, rtol=2e-2, atol=2e-2)
    gy = torch.randn_like(y)
    gx, gw, gb = torch.autograd.grad(y, [x, w, b], gy, create_graph=True)
    gxr, gwr, gbr = torch.autograd.grad(yr, [xr, wr, br], gy.float().cpu(),
                                        create_graph=True)
    torch.testing.assert_close(gx.float().cpu(), gxr, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(gw.float().cpu(), gwr, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(gb.float().cpu(), gbr, rtol=2e-2, atol=2e-2)
    # second order: grad of a grad-norm wrt the weight
    s = (gx.float() ** 2).sum()
    sw = torch.autograd.grad(s, w, retain_graph=True)[0]
    sr = torch.autograd.grad((gxr ** 2).sum(), wr, retain_graph=True)[0]
    torch.testing.assert_close(sw.cpu(), sr, rtol=5e-2, atol=5e-1)


def test_linear_deterministic_across_fresh_calls():
    torch.manual_seed(22)
    outs = []
    for trial in range(3):
        x = torch.randn(2, 9, 64, device=dev()).to(torch.bfloat16)
        torch.manual_seed(22)
        x = torch.randn(2, 9, 64, device=dev()).to(torch.bfloat16)
        w = torch.randn(2, 5, 64, device=dev())
        outs.append(ops.task_linear(x, w, None).cpu())
    assert (outs[0] == outs[1]).all().item() and (outs[1] == outs[2]).all().item()


def test_fp32_support_pass_tightens_meta_gradient():
    """--fp32_support_pass runs the inner-loop support chain in fp32 while
    targets stay bf16: the second-order meta-gradient must be at least as
    close to the fp32 oracle as the all-bf16 path (VERDICT r1 numerics
    lever)."""
    batch = _engine_batch()
    _, m_cpu = _engine_build(torch.device("cpu"))
    _, gc = _theta_grad(m_cpu, batch)
    _, m_bf = _engine_build(dev())
    _, g_bf = _theta_grad(m_bf, batch)
    _, m_mx = _engine_build(dev(), fp32_support="True")
    _, g_mx = _theta_grad(m_mx, batch)
    rel_bf = ((g_bf.cpu() - gc).norm() / gc.norm()).item()
    rel_mx = ((g_mx.cpu() - gc).norm() / gc.norm()).item()
    print(f"rel-L2 vs fp32 oracle: bf16={rel_bf:.4f} fp32-support={rel_mx:.4f}")
    assert rel_mx <= rel_bf * 1.05, (rel_mx, rel_bf)
    cos = torch.nn.functional.cosine_similarity(
        g_mx.cpu().flatten(), gc.flatten(), dim=0)
    assert cos > 0.97, f"cosine {cos:.5f}"
