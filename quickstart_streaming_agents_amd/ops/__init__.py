"""Op layer: the hand-written HIP/CDNA4 kernels (qsa_hip) + CPU references.

Fail-loud contract: on a GPU box the HIP extension MUST be importable —
there is no silent eager fallback for the hot ops.  CPU-only environments
(unit tests, this dev container) use ops.cpu_ref directly and never touch
the extension.
"""

from __future__ import annotations

_ext = None
_import_error: Exception | None = None
try:
    import torch  # noqa: F401  (loads libc10/libtorch the extension links)
    from .. import qsa_hip as _ext  # built in-tree by setup.py
except ImportError as e:  # pragma: no cover - exercised only when unbuilt
    _import_error = e


def have_ext() -> bool:
    return _ext is not None


def ext():
    """The qsa_hip module; raises loudly if the native build is missing."""
    if _ext is None:
        raise RuntimeError(
            "qsa_hip extension not built. Run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"(import error: {_import_error})")
    return _ext
