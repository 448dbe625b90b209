"""GPU tests for the MFMA conv trio (fwd / dgrad / wgrad) against the
fp32 torch reference, including the composed double-backward paths that
second-order MAML exercises."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from howtotrainyourmamlpytorch_amd import ops
from howtotrainyourmamlpytorch_amd.ops import reference as ref


def dev():
    return torch.device("cuda", 0)


def test_mfma_probe_layout():
    """Asymmetric-matrix check of the 16x16x32 bf16 fragment layout
    (guide rule G9: symmetric inputs cannot catch operand/output swaps)."""
    torch.manual_seed(0)
    A = torch.randn(16, 32, device=dev())
    B = torch.randn(32, 16, device=dev())
    (D,) = ops.hip_ext().mfma_probe(A, B)
    Dref = (A.to(torch.bfloat16).float() @ B.to(torch.bfloat16).float())
    torch.testing.assert_close(D, Dref, rtol=2e-2, atol=2e-2)
    # transpose-detecting: must NOT match the transposed product
    assert not torch.allclose(D, Dref.T, rtol=2e-2, atol=2e-1)


def _mk(T=2, NS=3, H=12, W=12, C=48, F=48, pad=1, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(T, NS, H, W, C, generator=g).to(dev(), torch.bfloat16)
    w = torch.randn(T, F, C, 3, 3, generator=g).mul(0.1).to(dev())
    b = torch.randn(T, F, generator=g).to(dev())
    return x, w, b


@pytest.mark.parametrize("C,F,pad", [(48, 48, 1), (64, 64, 1), (1, 64, 1),
                                     (3, 48, 1), (48, 48, 0)])
def test_tconv_fwd_matches_reference(C, F, pad):
    x, w, b = _mk(C=C, F=F, pad=pad)
    y = ops.task_conv3x3(x, w, b, stride=1, padding=pad)
    yr = ref.task_conv3x3(x.float().cpu(), w.cpu(), b.cpu(), stride=1, padding=pad)
    assert y.shape == yr.shape
    scale = yr.abs().max().item()
    torch.testing.assert_close(y.float().cpu(), yr, rtol=3e-2,
                               atol=3e-2 * max(scale, 1.0))


def test_tconv_first_order_grads_match_reference():
    x, w, b = _mk(C=48, F=48)
    x.requires_grad_(True); w.requires_grad_(True); b.requires_grad_(True)
    y = ops.task_conv3x3(x, w, b, stride=1, padding=1)
    gout = torch.randn_like(y)
    dx, dw, db = torch.autograd.grad(y, (x, w, b), gout)

    xr = x.detach().float().cpu().requires_grad_(True)
    wr = w.detach().cpu().requires_grad_(True)
    br = b.detach().cpu().requires_grad_(True)
    yr = ref.task_conv3x3(xr, wr, br, stride=1, padding=1)
    dxr, dwr, dbr = torch.autograd.grad(yr, (xr, wr, br), gout.float().cpu())

    for a, r, tol in ((dx.float().cpu(), dxr, 3e-2), (dw.cpu(), dwr, 3e-2),
                      (db.cpu(), dbr, 3e-2)):
        scale = r.abs().max().item()
        torch.testing.assert_close(a, r, rtol=5e-2, atol=tol * max(scale, 1.0))


def test_tconv_second_order_matches_reference():
    """The MAML pattern: grad of (a function of the first-order grads)."""
    x, w, b = _mk(T=2, NS=2, H=8, W=8, C=48, F=48, seed=3)
    w.requires_grad_(True); b.requires_grad_(True)

    def inner_outer(opmod, x_, w_, b_):
        y = opmod.task_conv3x3(x_, w_, b_, stride=1, padding=1)
        loss = (y.float() ** 2).mean()
        gw, = torch.autograd.grad(loss, (w_,), create_graph=True)
        w2 = w_ - 0.1 * gw
        y2 = opmod.task_conv3x3(x_, w2.to(w_.dtype), b_, stride=1, padding=1)
        outer = (y2.float() ** 2).mean()
        return torch.autograd.grad(outer, (w_, b_))

    gw, gb = inner_outer(ops, x, w, b)

    xr = x.detach().float().cpu()
    wr = w.detach().cpu().requires_grad_(True)
    br = b.detach().cpu().requires_grad_(True)
    gwr, gbr = inner_outer(ref, xr, wr, br)

    for a, r in ((gw.cpu(), gwr), (gb.cpu(), gbr)):
        scale = r.abs().max().item()
        torch.testing.assert_close(a, r, rtol=8e-2, atol=5e-2 * max(scale, 1e-3))


def test_tconv_wgrad_large_k_split():
    """K large enough to span multiple 4096-element K-chunks (atomics path)."""
    x, w, b = _mk(T=2, NS=6, H=32, W=32, C=48, F=48, seed=5)  # K = 6*32*32 = 6144
    x.requires_grad_(False); w.requires_grad_(True)
    y = ops.task_conv3x3(x, w, None, stride=1, padding=1)
    gout = torch.randn_like(y)
    (dw,) = torch.autograd.grad(y, (w,), gout)
    wr = w.detach().cpu().requires_grad_(True)
    yr = ref.task_conv3x3(x.detach().float().cpu(), wr, None, stride=1, padding=1)
    (dwr,) = torch.autograd.grad(yr, (wr,), gout.float().cpu())
    scale = dwr.abs().max().item()
    torch.testing.assert_close(dw.cpu(), dwr, rtol=5e-2, atol=3e-2 * max(scale, 1.0))


def test_conv_v2_bitwise_matches_v1():
    """The v2 async-pipelined kernel performs the identical MFMA sequence
    (same tiles, zero tails), so its output is bitwise equal to v1."""
    from howtotrainyourmamlpytorch_amd.ops import hip_ext
    ext = hip_ext()
    torch.manual_seed(11)
    for (T, NB, H, W, C, F, pad) in [(3, 7, 14, 14, 48, 48, 1),
                                     (2, 9, 28, 28, 64, 64, 1),
                                     (2, 5, 12, 12, 16, 48, 0),
                                     (1, 3, 10, 10, 8, 24, 1)]:
        x = torch.randn(T, NB, H, W, C, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(T, F, C, 3, 3, device="cuda")
        b = torch.randn(T, F, device="cuda")
        Ho, Wo = H + 2 * pad - 2, W + 2 * pad - 2
        wp1 = ext.tconv_repack(w, False)
        y1 = ext.tconv_mm(x, wp1, b, pad, Ho, Wo, False)[0]
        wp2 = ext.tconv_repack_v2(w, False)
        y2 = ext.tconv_mm_v2(x, wp2, b, pad, Ho, Wo, F, False)[0]
        assert (y1 == y2).all().item(), \
            f"v2 != v1 at {(T, NB, H, W, C, F, pad)}: " \
            f"max {(y1.float() - y2.float()).abs().max().item():e}"
        # dgrad orientation (flipped repack, pad' = 2 - pad)
        dy = torch.randn(T, NB, Ho, Wo, F, device="cuda", dtype=torch.bfloat16)
        wpd1 = ext.tconv_repack(w, True)
        dx1 = ext.tconv_mm(dy, wpd1, None, 2 - pad, H, W, False)[0]
        wpd2 = ext.tconv_repack_v2(w, True)
        dx2 = ext.tconv_mm_v2(dy, wpd2, None, 2 - pad, H, W, C, False)[0]
        assert (dx1 == dx2).all().item(), \
            f"v2 dgrad != v1 at {(T, NB, H, W, C, F, pad)}"


def test_wgrad_v2_matches_v1():
    """wgrad v2 (transposed operands + async staging) vs the v1 gather
    kernel: same math, different fp32 accumulation order -> tight fp
    tolerance, not bitwise.  Covers C in {1,3,48,64} incl. first-layer
    shapes v1 staged scalar."""
    from howtotrainyourmamlpytorch_amd.ops import hip_ext
    ext = hip_ext()
    torch.manual_seed(13)
    for (T, NB, H, W, C, F, pad) in [(3, 7, 14, 14, 48, 48, 1),
                                     (2, 9, 28, 28, 1, 48, 1),
                                     (2, 5, 28, 28, 3, 48, 1),
                                     (2, 6, 21, 21, 48, 48, 1),
                                     (2, 4, 28, 28, 64, 64, 1),
                                     (1, 3, 12, 12, 16, 32, 0)]:
        Ho, Wo = H + 2 * pad - 2, W + 2 * pad - 2
        x = torch.randn(T, NB, H, W, C, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn(T, NB, Ho, Wo, F, device="cuda", dtype=torch.bfloat16)
        dw1, db1 = ext.tconv_wgrad(dy, x, pad, True)
        dw2, db2 = ext.tconv_wgrad_v2(dy, x, pad, True)
        torch.testing.assert_close(dw2, dw1, rtol=1e-4, atol=1e-2), (T, NB, C)
        torch.testing.assert_close(db2, db1, rtol=1e-4, atol=1e-2)


def test_wgrad_v2_deterministic(monkeypatch):
    from howtotrainyourmamlpytorch_amd.ops import hip_ext
    ext = hip_ext()
    monkeypatch.setenv("MAML355_DETERMINISTIC", "1")
    torch.manual_seed(14)
    x = torch.randn(3, 9, 20, 20, 48, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(3, 9, 20, 20, 48, device="cuda", dtype=torch.bfloat16)
    a = ext.tconv_wgrad_v2(dy, x, 1, True)
    b = ext.tconv_wgrad_v2(dy, x, 1, True)
    assert (a[0] == b[0]).all().item() and (a[1] == b[1]).all().item()


def test_dconv_matches_gemm_path():
    """Direct small-C kernels vs the GEMM reference: fwd tight-bf16 equal,
    wgrad to fp accumulation tolerance."""
    from howtotrainyourmamlpytorch_amd.ops import hip_ext, reference as ref
    ext = hip_ext()
    torch.manual_seed(17)
    for (T, NB, H, W, C, F, pad) in [(2, 5, 28, 28, 1, 64, 1),
                                     (3, 7, 28, 28, 3, 48, 1),
                                     (2, 4, 12, 12, 3, 16, 1),
                                     (2, 3, 10, 10, 8, 32, 0)]:
        Ho, Wo = H + 2 * pad - 2, W + 2 * pad - 2
        x = torch.randn(T, NB, H, W, C, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(T, F, C, 3, 3, device="cuda")
        b = torch.randn(T, F, device="cuda")
        y = ext.dconv_fwd(x, w, b, pad, Ho, Wo)
        yr = ref.task_conv3x3(x.float().cpu(), w.cpu(), b.cpu(), 1, pad)
        torch.testing.assert_close(y.float().cpu(), yr, rtol=2e-2, atol=2e-2)
        dy = torch.randn(T, NB, Ho, Wo, F, device="cuda", dtype=torch.bfloat16)
        dw, db = ext.dconv_wgrad(dy, x, pad, True)
        dw1, db1 = ext.tconv_wgrad(dy, x, pad, True)
        torch.testing.assert_close(dw, dw1, rtol=1e-4, atol=1e-2)
        torch.testing.assert_close(db, db1, rtol=1e-4, atol=1e-2)


def test_dconv_wgrad_deterministic(monkeypatch):
    from howtotrainyourmamlpytorch_amd.ops import hip_ext
    ext = hip_ext()
    monkeypatch.setenv("MAML355_DETERMINISTIC", "1")
    torch.manual_seed(18)
    x = torch.randn(3, 9, 28, 28, 3, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(3, 9, 28, 28, 48, device="cuda", dtype=torch.bfloat16)
    a = ext.dconv_wgrad(dy, x, 1, True)
    b = ext.dconv_wgrad(dy, x, 1, True)
    assert (a[0] == b[0]).all().item() and (a[1] == b[1]).all().item()
