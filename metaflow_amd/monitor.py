"""System monitor + event logger.

Parity target: /root/reference/metaflow/monitor.py (measure/count context
managers emitting sidecar messages; NullMonitor default) and
/root/reference/metaflow/event_logger.py, with the reference's
`_system_monitor`/`_system_logger` singleton pattern
(system/system_monitor.py, system/system_logger.py).

Default is the no-op Null implementations; set ``MFX_MONITOR=debug`` /
``MFX_EVENT_LOGGER=debug`` to print to stderr, or ``=sidecar`` to emit
NDJSON through a ``monitor`` sidecar worker (lossy by design, like every
sidecar — see sidecar.py).
"""

import contextlib
import json
import os
import sys
import time


class NullMonitor(object):
    TYPE = "null"

    def init_environment(self):
        pass

    def terminate(self):
        pass

    @contextlib.contextmanager
    def measure(self, name):
        """Time a block; subclasses emit (name, elapsed_ms)."""
        yield

    @contextlib.contextmanager
    def count(self, name):
        """Count one occurrence of `name` when the block completes."""
        yield

    def gauge(self, name, value):
        pass


class DebugMonitor(NullMonitor):
    TYPE = "debug"

    @contextlib.contextmanager
    def measure(self, name):
        t0 = time.time()
        try:
            yield
        finally:
            sys.stderr.write("[mfx-monitor] measure %s: %.1f ms\n"
                             % (name, (time.time() - t0) * 1000))

    @contextlib.contextmanager
    def count(self, name):
        yield
        sys.stderr.write("[mfx-monitor] count %s +1\n" % name)

    def gauge(self, name, value):
        sys.stderr.write("[mfx-monitor] gauge %s=%s\n" % (name, value))


class SidecarMonitor(NullMonitor):
    """Emits measurements through the lossy sidecar channel."""

    TYPE = "sidecar"

    def __init__(self):
        self._sidecar = None

    def init_environment(self):
        from .sidecar import SidecarSubProcess

        out = os.environ.get("MFX_MONITOR_OUT",
                             "/tmp/mfx_monitor.jsonl")
        self._sidecar = SidecarSubProcess("monitor", {"out_path": out})

    def _send(self, kind, name, value=None):
        if self._sidecar is not None:
            self._sidecar.send(kind, {"name": name, "value": value,
                                      "ts": time.time()})

    @contextlib.contextmanager
    def measure(self, name):
        t0 = time.time()
        try:
            yield
        finally:
            self._send("measure", name, (time.time() - t0) * 1000)

    @contextlib.contextmanager
    def count(self, name):
        yield
        self._send("count", name, 1)

    def gauge(self, name, value):
        self._send("gauge", name, value)

    def terminate(self):
        if self._sidecar is not None:
            self._sidecar.terminate()
            self._sidecar = None


class NullEventLogger(object):
    TYPE = "null"

    def init_environment(self):
        pass

    def terminate(self):
        pass

    def log(self, payload):
        pass


class DebugEventLogger(NullEventLogger):
    TYPE = "debug"

    def log(self, payload):
        from .system_context import current_phase

        payload = dict(payload)
        payload.setdefault("phase", current_phase())
        sys.stderr.write("[mfx-event] %s\n" % json.dumps(payload))


MONITORS = {"null": NullMonitor, "debug": DebugMonitor,
            "sidecar": SidecarMonitor}
EVENT_LOGGERS = {"null": NullEventLogger, "debug": DebugEventLogger}

_system_monitor = None
_system_logger = None


def get_system_monitor():
    """Process-wide monitor singleton (reference: _system_monitor)."""
    global _system_monitor
    if _system_monitor is None:
        cls = MONITORS.get(os.environ.get("MFX_MONITOR", "null"),
                           NullMonitor)
        _system_monitor = cls()
        _system_monitor.init_environment()
    return _system_monitor


def get_system_logger():
    global _system_logger
    if _system_logger is None:
        cls = EVENT_LOGGERS.get(os.environ.get("MFX_EVENT_LOGGER", "null"),
                                NullEventLogger)
        _system_logger = cls()
        _system_logger.init_environment()
    return _system_logger
