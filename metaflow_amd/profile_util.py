"""Lightweight wall-clock profiling helpers.

Parity target: /root/reference/metaflow/metaflow_profile.py (from_start
ms checkpoints gated by an env var; profile() context manager).
"""

import contextlib
import os
import sys
import time

_T0 = time.time()
_ENABLED = bool(os.environ.get("MFX_PROFILE_FROM_START"))


def from_start(label):
    """Print ms-since-import when MFX_PROFILE_FROM_START is set."""
    if _ENABLED:
        sys.stderr.write("[mfx-profile] %8.1f ms  %s\n"
                         % ((time.time() - _T0) * 1000, label))


@contextlib.contextmanager
def profile(label, stream=None):
    t = time.time()
    yield
    out = stream or sys.stderr
    out.write("[mfx-profile] %s took %.1f ms\n"
              % (label, (time.time() - t) * 1000))
