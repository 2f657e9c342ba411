"""Artifact serializers: a priority-ordered, pluggable registry.

Parity target: /root/reference/metaflow/datastore/artifacts/serializer.py
(`SerializerStore` :149, `get_ordered_serializers` :200, lazy extension
`bootstrap` :252): subclassing ``ArtifactSerializer`` auto-registers via
``__init_subclass__`` (the metaclass-registration analog), serializers
are tried in ascending ``priority`` order, pickle is the universal
fallback, and the winning serializer's ``encoding`` is recorded per
artifact so deserialization is an exact registry lookup. Extensions
(``metaflow_amd_extensions`` namespace packages) contribute serializers
by exporting ``ARTIFACT_SERIALIZERS = [cls, ...]`` — loaded lazily on
first use so import order never matters.

MI355X addition: a zero-pickle tensor codec that writes dtype/shape
header + raw storage bytes, moving GPU tensors through a single D2H copy
instead of pickle's multiple buffer copies (SURVEY §2.2 note).
"""

import io
import pickle
import struct
import sys

ENC_PICKLE = "pickle-v4"
ENC_TENSOR = "tensor-v1"

_REGISTRY = {}          # encoding -> serializer instance
_ORDERED = None         # cache of priority-sorted instances
_bootstrapped = False


class ArtifactSerializer(object):
    """Base class; subclasses auto-register.

    Class attributes:
      encoding  — unique string recorded per artifact (required);
      priority  — ascending try-order (lower runs first; pickle is 1000).
    Methods: ``can_serialize(obj)``, ``serialize(obj) -> bytes``,
    ``deserialize(data) -> obj``.
    """

    encoding = None
    priority = 500

    def __init_subclass__(cls, **kwargs):
        super().__init_subclass__(**kwargs)
        if cls.encoding is not None:
            register_serializer(cls)

    def can_serialize(self, obj):
        raise NotImplementedError

    def serialize(self, obj):
        raise NotImplementedError

    def deserialize(self, data):
        raise NotImplementedError


def register_serializer(cls_or_instance):
    """Register a serializer (idempotent by encoding; later wins so an
    extension can override a built-in codec)."""
    global _ORDERED
    inst = (cls_or_instance() if isinstance(cls_or_instance, type)
            else cls_or_instance)
    if not inst.encoding:
        raise ValueError("serializer needs a non-empty `encoding`")
    _REGISTRY[inst.encoding] = inst
    _ORDERED = None
    return inst


def _bootstrap_extensions():
    """Lazily pull ARTIFACT_SERIALIZERS from extension packages
    (reference serializer.py:252 bootstrap; retried once per process)."""
    global _bootstrapped
    if _bootstrapped:
        return
    _bootstrapped = True
    try:
        from ..extension_support import iter_extension_modules
    except Exception:
        return
    for mod in iter_extension_modules():
        contrib = getattr(mod, "ARTIFACT_SERIALIZERS", None)
        if contrib is None and hasattr(mod, "get_plugins"):
            try:
                contrib = mod.get_plugins().get("artifact_serializers")
            except Exception:
                contrib = None
        for cls in contrib or ():
            try:
                register_serializer(cls)
            except Exception as ex:  # never kill the engine
                sys.stderr.write(
                    "[mfx] warning: serializer from %s skipped: %r\n"
                    % (getattr(mod, "__name__", mod), ex))


def get_ordered_serializers():
    """Registered serializers in ascending priority order."""
    global _ORDERED
    _bootstrap_extensions()
    if _ORDERED is None:
        _ORDERED = sorted(_REGISTRY.values(), key=lambda s: s.priority)
    return _ORDERED


# ============================ tensor codec =================================
_TENSOR_MAGIC = b"MFXT\x01"

# torch dtype <-> wire id (append-only; never renumber)
_DTYPE_IDS = {}
_DTYPE_FROM_ID = {}


def _init_dtype_table():
    import torch

    table = [
        torch.float32, torch.float64, torch.float16, torch.bfloat16,
        torch.int8, torch.uint8, torch.int16, torch.int32, torch.int64,
        torch.bool,
    ]
    if hasattr(torch, "float8_e4m3fn"):
        table.append(torch.float8_e4m3fn)
    if hasattr(torch, "float8_e5m2"):
        table.append(torch.float8_e5m2)
    for i, dt in enumerate(table):
        _DTYPE_IDS[dt] = i
        _DTYPE_FROM_ID[i] = dt


class TensorSerializer(ArtifactSerializer):
    """dtype/shape header + raw contiguous bytes; GPU tensors staged to
    CPU with one copy (pinned staging lives in the checkpoint path)."""

    encoding = ENC_TENSOR
    priority = 50

    def can_serialize(self, obj):
        torch = sys.modules.get("torch")
        return torch is not None and isinstance(obj, torch.Tensor)

    def serialize(self, obj):
        import torch

        if not _DTYPE_IDS:
            _init_dtype_table()
        t = obj.detach()
        if t.is_cuda:
            t = t.to("cpu")
        t = t.contiguous()
        shape = tuple(t.shape)
        header = _TENSOR_MAGIC + struct.pack(
            "<BB", _DTYPE_IDS[t.dtype], len(shape))
        header += struct.pack("<%dq" % len(shape), *shape)
        raw = t.reshape(-1).view(torch.uint8).numpy().tobytes()
        return header + raw

    def deserialize(self, data):
        import torch

        if not _DTYPE_FROM_ID:
            _init_dtype_table()
        assert data[:5] == _TENSOR_MAGIC, "bad tensor blob"
        dtype_id, ndim = struct.unpack_from("<BB", data, 5)
        shape = struct.unpack_from("<%dq" % ndim, data, 7)
        offset = 7 + 8 * ndim
        dtype = _DTYPE_FROM_ID[dtype_id]
        t = torch.frombuffer(bytearray(data[offset:]), dtype=torch.uint8)
        return t.view(dtype).reshape(shape)


class NumpySerializer(ArtifactSerializer):
    """Zero-pickle numpy codec: dtype/shape header + raw buffer bytes.

    pickle round-trips a 1 GiB uint8 array in ~2.6 s (dumps) + ~1.0 s
    (loads) of pure buffer shuffling — the dominant per-task cost of
    BASELINE config 2 (profiles/bench_results_r02.md). This codec
    serializes with a single memcpy and deserializes ZERO-copy:
    ``np.frombuffer`` over the CAS bytes, so the returned array is
    READ-ONLY (content-addressed artifacts are immutable by design;
    call ``.copy()`` for a mutable scratch array). Object dtypes fall
    through to pickle."""

    encoding = "numpy-v1"
    priority = 60
    _MAGIC = b"MFXN1"

    def can_serialize(self, obj):
        np = sys.modules.get("numpy")
        return (np is not None and isinstance(obj, np.ndarray)
                and obj.dtype.hasobject is False)

    def serialize(self, obj):
        import numpy as np

        # (ascontiguousarray promotes 0-d to 1-d; 0-d is contiguous)
        arr = obj if obj.flags.c_contiguous \
            else np.ascontiguousarray(obj)
        dt = arr.dtype.str.encode()          # e.g. b"<u1", b"<f4"
        header = self._MAGIC + struct.pack("<BB", len(dt), arr.ndim)
        header += dt
        header += struct.pack("<%dq" % arr.ndim, *arr.shape)
        return header + arr.tobytes()

    def deserialize(self, data):
        import numpy as np

        assert data[:5] == self._MAGIC, "bad numpy blob"
        dtlen, ndim = struct.unpack_from("<BB", data, 5)
        dt = data[7:7 + dtlen].decode()
        shape = struct.unpack_from("<%dq" % ndim, data, 7 + dtlen)
        offset = 7 + dtlen + 8 * ndim
        return np.frombuffer(data, dtype=np.dtype(dt),
                             offset=offset).reshape(shape)


class PickleSerializer(ArtifactSerializer):
    """Universal fallback (reference registers it last,
    plugins/__init__.py:192)."""

    encoding = ENC_PICKLE
    priority = 1000

    def can_serialize(self, obj):
        return True

    def serialize(self, obj):
        buf = io.BytesIO()
        pickle.dump(obj, buf, protocol=4)
        return buf.getvalue()

    def deserialize(self, data):
        return pickle.loads(data)


# module-level convenience API (what task_datastore uses)
def serialize(obj):
    """Returns (bytes, encoding) from the first willing serializer."""
    for s in get_ordered_serializers():
        if s.can_serialize(obj):
            try:
                return s.serialize(obj), s.encoding
            except Exception:
                if s.encoding == ENC_PICKLE:
                    raise
                continue  # fall through to lower-priority codecs
    raise ValueError("no serializer accepted %r" % type(obj))


def deserialize(data, encoding):
    s = _REGISTRY.get(encoding)
    if s is None:
        _bootstrap_extensions()
        s = _REGISTRY.get(encoding)
    if s is None and encoding.startswith("pickle"):
        s = _REGISTRY.get(ENC_PICKLE)
    if s is None:
        raise ValueError("Unknown artifact encoding %r (no registered "
                         "serializer)" % encoding)
    return s.deserialize(data)


def serialize_tensor(t):
    """Back-compat helper (tensor codec bytes without the registry)."""
    return _REGISTRY[ENC_TENSOR].serialize(t)


def deserialize_tensor(data):
    return _REGISTRY[ENC_TENSOR].deserialize(data)


def type_name(obj):
    t = type(obj)
    mod = getattr(t, "__module__", "")
    return "%s.%s" % (mod, t.__name__) if mod else t.__name__
