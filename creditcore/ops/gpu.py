"""Loader for the in-tree HIP extension (csrc/ -> creditcore/_ccore.so).

The extension is built for gfx950 by ``python setup.py build_ext --inplace``
or ``__graft_entry__.build()``; the resulting .so lives inside the package so
it travels with the repo snapshot. On a machine with a visible GPU the
extension is REQUIRED: scoring ops raise ExtensionMissing rather than
silently falling back to an eager/CPU path (the CPU path must be requested
explicitly via device="cpu").
"""

from __future__ import annotations

import importlib

_ext = None
_err: Exception | None = None


class ExtensionMissing(RuntimeError):
    pass


def _try_load():
    global _ext, _err
    if _ext is not None or _err is not None:
        return
    try:
        import torch  # noqa: F401  (the extension links against libtorch)

        _ext = importlib.import_module("creditcore._ccore")
    except Exception as e:  # pragma: no cover
        _err = e


def available() -> bool:
    _try_load()
    return _ext is not None


def ext():
    """Return the extension module; raise loudly if missing."""
    _try_load()
    if _ext is None:
        raise ExtensionMissing(
            "creditcore._ccore HIP extension not built/loadable "
            f"(build with `python setup.py build_ext --inplace`): {_err}"
        )
    return _ext
