"""maggy_amd.ops — hand-written CDNA4 HIP kernels + python wrappers.

Loading policy (no silent eager fallback): on a machine with a GPU the
compiled extension MUST load — ``require_ext()`` raises
HipExtensionMissingError otherwise.  On CPU-only hosts (this dev container,
CI) the wrappers fall back to torch reference implementations so the
scheduler/optimizer logic stays testable.
"""
import importlib.util
import os

from maggy_amd.exceptions import HipExtensionMissingError

_ext = None
_ext_error = None


def _try_load():
    global _ext, _ext_error
    if _ext is not None:
        return _ext
    from maggy_amd.ops.build import SO_PATH

    if not os.path.exists(SO_PATH):
        _ext_error = "extension not built ({} missing)".format(SO_PATH)
        return None
    try:
        spec = importlib.util.spec_from_file_location("_maggy_hip", SO_PATH)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
        return _ext
    except Exception as e:  # keep the reason for require_ext()
        _ext_error = repr(e)
        return None


def ext_or_none():
    return _try_load()


def require_ext():
    """Return the extension module; raise loudly if a GPU is present and it
    cannot be loaded."""
    mod = _try_load()
    if mod is None:
        raise HipExtensionMissingError(_ext_error or "")
    return mod


def has_ext():
    return _try_load() is not None


from maggy_amd.ops.fused_adam import FusedAdam, FusedSGD  # noqa: E402,F401
from maggy_amd.ops.reduce import grad_l2norm, metric_mean, metric_sum  # noqa: E402,F401

__all__ = [
    "FusedAdam", "FusedSGD", "grad_l2norm", "metric_mean", "metric_sum",
    "require_ext", "ext_or_none", "has_ext",
]
