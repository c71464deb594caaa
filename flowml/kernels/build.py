"""Build the flowml HIP extension in-tree for gfx950.

Usage: ``python -m flowml.kernels.build``  (also called by __graft_entry__.build).

Uses torch.utils.cpp_extension (which drives hipcc for the .hip TUs under
PYTORCH_ROCM_ARCH=gfx950) with an in-tree build directory, then copies the
resulting .so to flowml/kernels/_flowml_hip.so so it ships to the GPU box
with the repo snapshot (a JIT cache under ~/.cache would not).
"""

from __future__ import annotations

import os
import shutil
import sys

_HERE = os.path.dirname(os.path.abspath(__file__))


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils import cpp_extension

    build_dir = os.path.join(_HERE, "_build")
    os.makedirs(build_dir, exist_ok=True)
    sources = [
        os.path.join(_HERE, "ext.cpp"),
        os.path.join(_HERE, "tree_kernels.hip"),
        os.path.join(_HERE, "mlp_kernels.hip"),
    ]
    cflags = ["-O3"]
    hipflags = ["-O3", "--offload-arch=gfx950"]
    if os.environ.get("FLOWML_ASAN", "0") == "1":
        # device+host address sanitizer build: run the GPU
        # test suite against this to catch OOB in the atomic kernels
        cflags += ["-fsanitize=address", "-shared-libasan"]
        hipflags += ["-fsanitize=address", "-shared-libasan"]
    mod_path = cpp_extension.load(
        name="_flowml_hip",
        sources=sources,
        build_directory=build_dir,
        extra_cflags=cflags,
        extra_cuda_cflags=hipflags,
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
    )
    so_src = os.path.join(build_dir, "_flowml_hip.so")
    if not os.path.exists(so_src):
        raise RuntimeError(f"build produced no {so_src}")
    so_dst = os.path.join(_HERE, "_flowml_hip.so")
    shutil.copy2(so_src, so_dst)
    return so_dst


if __name__ == "__main__":
    path = build()
    print(f"built {path}")
    sys.exit(0)
