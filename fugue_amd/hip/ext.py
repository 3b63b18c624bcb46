"""Build/load the CDNA4 relational kernel extension.

The extension is built IN-TREE (``fugue_amd/hip/_build``) so the ``.so``
travels with the repo snapshot to GPU machines.  ``PYTORCH_ROCM_ARCH`` is
pinned to gfx950 (MI355X) — no multi-arch fat binaries.
"""
import os
import threading
from typing import Any, Optional

_BUILD_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_build")
_LOCK = threading.Lock()
_EXT: Optional[Any] = None
_EXT_ERROR: Optional[Exception] = None


def build_extension(verbose: bool = False) -> Any:
    """Compile (if needed) and load the extension module."""
    global _EXT, _EXT_ERROR
    with _LOCK:
        if _EXT is not None:
            return _EXT
        os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
        os.makedirs(_BUILD_DIR, exist_ok=True)
        from torch.utils.cpp_extension import load

        src_dir = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
        _EXT = load(
            name="fugue_amd_hip",
            sources=[
                os.path.join(src_dir, "bindings.cpp"),
                os.path.join(src_dir, "relational.hip"),
            ],
            build_directory=_BUILD_DIR,
            extra_cflags=["-O3"],
            extra_cuda_cflags=["-O3"],
            verbose=verbose,
        )
        return _EXT


def get_ext() -> Any:
    """The loaded extension; raises loudly if unavailable on a GPU host.

    The HIP engine must never fall back silently to an eager path on a GPU
    machine — a missing native extension is an error there.
    """
    global _EXT, _EXT_ERROR
    if _EXT is not None:
        return _EXT
    if _EXT_ERROR is not None:
        raise RuntimeError(
            f"fugue_amd HIP extension failed to build: {_EXT_ERROR}"
        ) from _EXT_ERROR
    try:
        return build_extension()
    except Exception as e:  # pragma: no cover
        _EXT_ERROR = e
        raise RuntimeError(
            f"fugue_amd HIP extension failed to build: {e}"
        ) from e


def ext_available() -> bool:
    try:
        get_ext()
        return True
    except Exception:
        return False
