"""In-tree build of the anovos_amd HIP extension for gfx950.

Invoked by __graft_entry__.build() (the driver's does-it-build check, run
on CPU — hipcc cross-compiles gfx950 without a GPU) and manually. The
built .so lands next to the sources so it travels to the GPU box with
the repo snapshot (JIT caches under ~/.cache do not).
"""

from __future__ import annotations

import os
import shutil

HERE = os.path.dirname(os.path.abspath(__file__))


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load

    build_dir = os.path.join(HERE, "build")
    os.makedirs(build_dir, exist_ok=True)
    mod = load(
        name="anovos_hip",
        sources=[
            os.path.join(HERE, "anovos_bindings.hip"),
            os.path.join(HERE, "anovos_kernels.hip"),
            os.path.join(HERE, "corr_mfma.hip"),
        ],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        build_directory=build_dir,
        verbose=verbose,
        is_python_module=False,  # just build; we place and load the .so ourselves
    )
    so = os.path.join(build_dir, "anovos_hip.so")
    dst = os.path.join(HERE, "anovos_hip.so")
    if os.path.exists(so):
        shutil.copy2(so, dst)
    return dst


if __name__ == "__main__":
    print(build())
