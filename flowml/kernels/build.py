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
    asan = os.environ.get("FLOWML_ASAN", "0") == "1"
    # force (not setdefault): an inherited PYTORCH_ROCM_ARCH=gfx950 would
    # conflict with the xnack+ objects the device sanitizer needs
    os.environ["PYTORCH_ROCM_ARCH"] = "gfx950:xnack+" if asan else "gfx950"
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
    if asan:
        # DEVICE address-sanitizer build: catches OOB in the
        # atomic kernels. Device ASAN needs xnack+ code objects
        # (PYTORCH_ROCM_ARCH=gfx950:xnack+ above); the host TU is left
        # un-instrumented (g++ lacks clang's -shared-libasan, and the
        # kernels are the risk surface).
        hipflags = [f.replace("--offload-arch=gfx950",
                              "--offload-arch=gfx950:xnack+") for f in hipflags]
        hipflags += ["-fsanitize=address", "-shared-libasan"]
    try:
        mod_path = _load(cpp_extension, sources, build_dir, cflags, hipflags,
                         verbose)
    except OSError as e:
        if asan:
            raise RuntimeError(
                "ASAN build compiled but needs the sanitizer runtime "
                "preloaded to import. Run under:\n  LD_PRELOAD=$("
                "/opt/rocm/lib/llvm/bin/clang -print-file-name="
                "libclang_rt.asan-x86_64.so) ASAN_OPTIONS=detect_leaks=0 "
                "HSA_XNACK=1 python ..."
            ) from e
        raise
    so_path = os.path.join(build_dir, "_flowml_hip.so")
    dst = os.path.join(_HERE, "_flowml_hip.so")
    shutil.copy2(so_path, dst)
    if verbose:
        print(f"built {dst}")
    return mod_path


def _load(cpp_extension, sources, build_dir, cflags, hipflags, verbose):
    return cpp_extension.load(
        name="_flowml_hip",
        sources=sources,
        build_directory=build_dir,
        extra_cflags=cflags,
        extra_cuda_cflags=hipflags,
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
    )


if __name__ == "__main__":
    build()
    sys.exit(0)
