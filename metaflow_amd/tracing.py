"""Optional OpenTelemetry tracing with no-op stubs.

Parity target: /root/reference/metaflow/tracing/ — spans activate only when
MFX_OTEL_ENDPOINT is configured AND opentelemetry is importable; otherwise
everything is a zero-cost no-op. Trace context propagates into child task
processes via env (inject_tracing_vars, reference runtime.py:2337).
"""

import contextlib
import os

_TRACING = None


def _init():
    global _TRACING
    if _TRACING is not None:
        return _TRACING
    endpoint = os.environ.get("MFX_OTEL_ENDPOINT")
    if not endpoint:
        _TRACING = False
        return False
    try:
        from opentelemetry import trace
        from opentelemetry.exporter.otlp.proto.grpc.trace_exporter import (
            OTLPSpanExporter,
        )
        from opentelemetry.sdk.trace import TracerProvider
        from opentelemetry.sdk.trace.export import BatchSpanProcessor

        provider = TracerProvider()
        provider.add_span_processor(
            BatchSpanProcessor(OTLPSpanExporter(endpoint=endpoint)))
        trace.set_tracer_provider(provider)
        _TRACING = trace.get_tracer("metaflow_amd")
    except Exception:
        _TRACING = False
    return _TRACING


@contextlib.contextmanager
def span(name, attributes=None):
    tracer = _init()
    if not tracer:
        yield None
        return
    with tracer.start_as_current_span(name) as s:
        for k, v in (attributes or {}).items():
            s.set_attribute(k, v)
        yield s


def cli(name):
    """Decorator putting a span around a CLI entry point."""

    def deco(f):
        def wrapped(*args, **kwargs):
            with span(name):
                return f(*args, **kwargs)

        wrapped.__name__ = getattr(f, "__name__", "wrapped")
        return wrapped

    return deco


def inject_tracing_vars(env):
    """Propagate trace context into a child process env dict."""
    if not _init():
        return env
    try:
        from opentelemetry.propagate import inject

        carrier = {}
        inject(carrier)
        env.update({"TRACEPARENT_%s" % k.upper(): v
                    for k, v in carrier.items()})
    except Exception:
        pass
    return env
