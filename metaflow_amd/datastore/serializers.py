"""Artifact serializers.

Parity target: /root/reference/metaflow/datastore/artifacts/serializer.py
(priority-ordered registry, pickle as universal fallback, per-artifact
encoding recorded). MI355X addition: a zero-pickle tensor codec that writes
dtype/shape header + raw storage bytes, moving GPU tensors through a single
D2H copy instead of pickle's multiple buffer copies (SURVEY §2.2 note).
"""

import io
import pickle
import struct

ENC_PICKLE = "pickle-v4"
ENC_TENSOR = "tensor-v1"

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


def _is_tensor(obj):
    try:
        import sys

        torch = sys.modules.get("torch")
        if torch is None:
            return False
        return isinstance(obj, torch.Tensor)
    except Exception:
        return False


def serialize_tensor(t):
    """dtype/shape header + raw contiguous bytes. GPU tensors are staged to
    CPU with a single non-blocking copy (pinned staging happens in the
    checkpoint path; here we accept a plain copy)."""
    import torch

    if not _DTYPE_IDS:
        _init_dtype_table()
    t = t.detach()
    if t.is_cuda:
        t = t.to("cpu")
    t = t.contiguous()
    shape = tuple(t.shape)
    header = _TENSOR_MAGIC + struct.pack(
        "<BB", _DTYPE_IDS[t.dtype], len(shape))
    header += struct.pack("<%dq" % len(shape), *shape)
    raw = t.reshape(-1).view(torch.uint8).numpy().tobytes()
    return header + raw


def deserialize_tensor(data):
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


def serialize(obj):
    """Returns (bytes, encoding)."""
    if _is_tensor(obj):
        try:
            return serialize_tensor(obj), ENC_TENSOR
        except Exception:
            pass
    buf = io.BytesIO()
    pickle.dump(obj, buf, protocol=4)
    return buf.getvalue(), ENC_PICKLE


def deserialize(data, encoding):
    if encoding == ENC_TENSOR:
        return deserialize_tensor(data)
    if encoding.startswith("pickle"):
        return pickle.loads(data)
    raise ValueError("Unknown artifact encoding %r" % encoding)


def type_name(obj):
    t = type(obj)
    mod = getattr(t, "__module__", "")
    return "%s.%s" % (mod, t.__name__) if mod else t.__name__
