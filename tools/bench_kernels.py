"""Isolated kernel micro-benchmarks on flagship shapes (run on MI355X):

    python tools/bench_kernels.py

Prints per-kernel time, effective TFLOP/s (conv) or TB/s (memory-bound),
so optimization effort goes where the roofline says.
"""

import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from howtotrainyourmamlpytorch_amd import ops  # noqa: E402
from howtotrainyourmamlpytorch_amd.ops import hip_ext  # noqa: E402


def timeit(fn, reps=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def bench_conv(T, NB, H, W, C, F, tag):
    dev = torch.device("cuda")
    x = torch.randn(T, NB, H, W, C, device=dev).to(torch.bfloat16)
    w = torch.randn(T, F, C, 3, 3, device=dev)
    b = torch.randn(T, F, device=dev)
    ext = hip_ext()
    wp = ext.tconv_repack(w, False)
    wpd = ext.tconv_repack(w, True)
    y = ext.tconv_mm(x, wp, b, 1, H, W, False)[0]
    dy = torch.randn_like(y)

    flops = 2.0 * T * NB * H * W * F * 9 * C
    t_f = timeit(lambda: ext.tconv_mm(x, wp, b, 1, H, W, False))
    t_d = timeit(lambda: ext.tconv_mm(dy, wpd, None, 1, H, W, False))
    t_w = timeit(lambda: ext.tconv_wgrad(dy, x, 1, True))
    print(f"[conv {tag}] T{T} NB{NB} {H}x{W} C{C}->F{F}  "
          f"fwd {t_f*1e6:7.1f}us {flops/t_f/1e12:6.1f}TF | "
          f"dgrad {t_d*1e6:7.1f}us {flops/t_d/1e12:6.1f}TF | "
          f"wgrad {t_w*1e6:7.1f}us {flops/t_w/1e12:6.1f}TF")
    if C % 8 == 0:
        wp2 = ext.tconv_repack_v2(w, False)
        wpd2 = ext.tconv_repack_v2(w, True)
        t_f2 = timeit(lambda: ext.tconv_mm_v2(x, wp2, b, 1, H, W, F, False))
        t_d2 = timeit(lambda: ext.tconv_mm_v2(dy, wpd2, None, 1, H, W, C, False))
        print(f"[conv {tag}]   v2:                 "
              f"fwd {t_f2*1e6:7.1f}us {flops/t_f2/1e12:6.1f}TF | "
              f"dgrad {t_d2*1e6:7.1f}us {flops/t_d2/1e12:6.1f}TF")
    if F <= 64:
        t_w2 = timeit(lambda: ext.tconv_wgrad_v2(dy, x, 1, True))
        print(f"[conv {tag}]   wgrad_v2:           "
              f"{t_w2*1e6:7.1f}us {flops/t_w2/1e12:6.1f}TF")


def bench_bn(T, M, C, tag):
    dev = torch.device("cuda")
    x = torch.randn(T, M, C, device=dev).to(torch.bfloat16)
    gamma = torch.rand(C, device=dev) + 0.5
    beta = torch.randn(C, device=dev)
    ext = hip_ext()
    y, mean, var, rstd = ext.bn_act_fwd(x, gamma, beta, 1e-5, 0.01, True)
    dy = torch.randn_like(y)
    nbytes = T * M * C * 2
    t_f = timeit(lambda: ext.bn_act_fwd(x, gamma, beta, 1e-5, 0.01, True))
    t_b = timeit(lambda: ext.bn_act_bwd(dy, x, mean, rstd, gamma, beta, 0.01, True))
    # fwd: read x twice (sums+norm) + write y; bwd: read dy,x twice + write dx
    print(f"[bn {tag}] T{T} M{M} C{C}  fwd {t_f*1e6:6.1f}us "
          f"{3*nbytes/t_f/1e12:5.2f}TB/s | bwd {t_b*1e6:6.1f}us "
          f"{5*nbytes/t_b/1e12:5.2f}TB/s")


def bench_pool(N, H, W, C, tag):
    dev = torch.device("cuda")
    x = torch.randn(N, H, W, C, device=dev).to(torch.bfloat16)
    ext = hip_ext()
    y, mask = ext.maxpool2x2_fwd(x)
    dy = torch.randn_like(y)
    nbytes = N * H * W * C * 2
    t_f = timeit(lambda: ext.maxpool2x2_fwd(x))
    t_b = timeit(lambda: ext.maxpool2x2_bwd(dy, mask, H, W))
    print(f"[pool {tag}] N{N} {H}x{W} C{C}  fwd {t_f*1e6:6.1f}us "
          f"{1.625*nbytes/t_f/1e12:5.2f}TB/s | bwd {t_b*1e6:6.1f}us "
          f"{1.875*nbytes/t_b/1e12:5.2f}TB/s")


def main():
    assert torch.cuda.is_available()
    print(torch.cuda.get_device_name(0))
    # mini-imagenet flagship: 8 tasks/GPU, target pass (75 imgs/task)
    bench_conv(8, 75, 84, 84, 3, 48, "mi-conv0-tgt")
    bench_conv(8, 75, 42, 42, 48, 48, "mi-conv1-tgt")
    bench_conv(8, 75, 21, 21, 48, 48, "mi-conv2-tgt")
    bench_conv(8, 5, 42, 42, 48, 48, "mi-conv1-sup")
    # omniglot 20w5s: 100-image support
    bench_conv(8, 100, 28, 28, 64, 64, "om-conv1-sup")
    bench_bn(8, 75 * 42 * 42, 48, "mi-bn1-tgt")
    bench_bn(8, 100 * 28 * 28, 64, "om-bn1")
    bench_pool(8 * 75, 84, 84, 48, "mi-pool0-tgt")
    bench_conv0_wrapper(8, 75, 84, 84, 3, 48, "mi-conv0-tgt")
    bench_conv0_wrapper(8, 100, 28, 28, 1, 64, "om-conv0-sup")




def bench_conv0_wrapper(T, NB, H, W, C, F, tag):
    """First-layer conv through the dispatch wrapper (exercises the
    channel-pad-to-8 path): fwd and fwd+wgrad."""
    dev = torch.device("cuda")
    x = torch.randn(T, NB, H, W, C, device=dev).to(torch.bfloat16)
    w = torch.randn(T, F, C, 3, 3, device=dev, requires_grad=True)
    b = torch.randn(T, F, device=dev, requires_grad=True)
    flops = 2.0 * T * NB * H * W * F * 9 * C
    t_f = timeit(lambda: ops.task_conv3x3(x, w, b), reps=20)

    def fb():
        y = ops.task_conv3x3(x, w, b)
        torch.autograd.grad(y.float().sum(), [w, b])
    t_fb = timeit(fb, reps=20)
    print(f"[conv0w {tag}] T{T} NB{NB} {H}x{W} C{C}->F{F}  "
          f"fwd {t_f*1e6:7.1f}us {flops/t_f/1e12:6.1f}TF | "
          f"fwd+wgrad {t_fb*1e6:7.1f}us")


if __name__ == "__main__":
    main()
