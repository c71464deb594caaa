"""HIP/CDNA4 kernel extension loader.

Kernels live in ``flowml/kernels/*.hip`` and are compiled in-tree to
``flowml/kernels/_flowml_hip.so`` for gfx950 (``python -m flowml.kernels.build``
or ``__graft_entry__.build()``); the built .so ships to the GPU box with the
repo snapshot. The loader never JIT-compiles at import time.
"""

from __future__ import annotations

import os

_HERE = os.path.dirname(os.path.abspath(__file__))
SO_NAME = "_flowml_hip"


def so_path() -> str:
    return os.path.join(_HERE, f"{SO_NAME}.so")


def load_extension():
    """Import the compiled extension module (torch must be imported first)."""
    import importlib.util
    import torch  # noqa: F401  (registers the c10/torch symbols the .so needs)

    path = so_path()
    if not os.path.exists(path):
        raise FileNotFoundError(
            f"{path} not built; run `python -m flowml.kernels.build`"
        )
    spec = importlib.util.spec_from_file_location(SO_NAME, path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod
