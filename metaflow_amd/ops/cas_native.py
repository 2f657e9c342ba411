"""Loader for the native C++ CAS engine (_mfx_cas extension).

Returns None-ish (raises) when the extension isn't built; cas.py treats any
failure here as "use the pure-Python path". Unlike the GPU kernels, the CAS
engine is an optional accelerator — correctness is identical either way.
"""

_engine = None
_tried = False


def engine():
    global _engine, _tried
    if _engine is None:
        import sys

        if "torch" not in sys.modules:
            # the extension links torch's libs, so first use would pull
            # in the FULL torch import (~1.5 s) — that cost made every
            # small-artifact task subprocess 10x slower (bench_flow
            # 0.47 s -> 5.3 s). Processes that already use torch
            # (training, checkpoint, datatools on big tensors) get the
            # native engine; everyone else keeps the plain-python path.
            raise ImportError(
                "native CAS engine deferred: torch not imported")
    if not _tried:
        _tried = True
        try:
            import torch  # noqa: F401  (loads libc10.so for the extension)

            from . import _mfx_cas  # built in-tree by setup.py

            _engine = _mfx_cas.Engine()
        except Exception:
            _engine = None
    if _engine is None:
        raise ImportError("_mfx_cas extension not built")
    return _engine
