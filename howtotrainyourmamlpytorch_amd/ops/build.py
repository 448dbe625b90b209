"""In-tree build of the gfx950 HIP extension.

Builds ``_maml_hip.so`` next to this file (NOT into a JIT cache under
~/.cache — the in-tree .so travels to the GPU box with the repo snapshot).

Run: ``python -m howtotrainyourmamlpytorch_amd.ops.build``
"""

from __future__ import annotations

import os
import shutil

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(PKG_DIR, "hip")
SO_NAME = "_maml_hip"

SOURCES = [
    os.path.join(HIP_DIR, "bindings.cpp"),
    os.path.join(HIP_DIR, "bn_act.hip"),
    os.path.join(HIP_DIR, "pool.hip"),
    os.path.join(HIP_DIR, "softmax_ce.hip"),
    os.path.join(HIP_DIR, "lslr.hip"),
    os.path.join(HIP_DIR, "tconv.hip"),
    os.path.join(HIP_DIR, "bn_dbwd.hip"),
    os.path.join(HIP_DIR, "adam.hip"),
    os.path.join(HIP_DIR, "dconv.hip"),
    os.path.join(HIP_DIR, "linear.hip"),
]


def build_extension(verbose: bool = False) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load

    sources = [s for s in SOURCES if os.path.isfile(s)]
    build_dir = os.path.join(PKG_DIR, "_build")
    os.makedirs(build_dir, exist_ok=True)
    module = load(
        name=SO_NAME,
        sources=sources,
        build_directory=build_dir,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17", "--offload-arch=gfx950"],
        verbose=verbose,
        is_python_module=False,  # just build; we dlopen from the copied path
    )
    built = os.path.join(build_dir, SO_NAME + ".so")
    dest = os.path.join(PKG_DIR, SO_NAME + ".so")
    shutil.copy2(built, dest)
    if verbose:
        print(f"built {dest}")
    return dest


if __name__ == "__main__":
    build_extension(verbose=True)
