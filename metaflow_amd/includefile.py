"""IncludeFile: a file-typed parameter whose content is uploaded to the
content-addressed store at run start and lazily decoded in steps.

Parity target: /root/reference/metaflow/includefile.py:234,386
(IncludeFile + UploaderV1/V2): the parameter ARTIFACT is a small
IncludedFile handle ({key, size, ...}); the bytes live as one raw CAS
blob — a 1 GB include is uploaded once (CAS dedup across runs), the
handle propagates via artifact passdown, and only steps that actually
ACCESS the parameter download/decode it (FlowSpec.__getattr__ hook).
"""

import os

from .parameters import Parameter


class IncludedFile(object):
    """Lazy handle stored as the parameter's artifact value."""

    def __init__(self, descriptor):
        # descriptor: {"key", "size", "is_text", "encoding"} (CAS blob),
        # or legacy {"data": ...} / {"path": ...}
        self.descriptor = descriptor

    @property
    def size(self):
        return self.descriptor.get("size")

    def decode(self, task_datastore=None):
        d = self.descriptor
        if "data" in d:
            return d["data"]
        if "path" in d:
            with open(d["path"], "rb") as f:
                data = f.read()
        elif "key" in d:
            if task_datastore is None:
                raise ValueError(
                    "IncludedFile %s needs a datastore to decode"
                    % d["key"][:16])
            [(_k, data)] = list(
                task_datastore._ca_store.load_blobs([d["key"]]))
        else:
            raise ValueError("IncludedFile has no content")
        if d.get("is_text"):
            return bytes(data).decode(d.get("encoding") or "utf-8")
        return bytes(data)

    def __repr__(self):
        d = dict(self.descriptor)
        if "data" in d:
            d["data"] = "<%d chars>" % len(d["data"])
        return "IncludedFile(%s)" % d


class _DelayedUpload(object):
    """convert() result before the run-start CAS upload: carries the
    local path so the (possibly huge) file is read exactly once, by the
    uploader, not by parameter parsing."""

    def __init__(self, path, is_text, encoding):
        self.path = path
        self.is_text = is_text
        self.encoding = encoding


def upload_include_files(values, flow_datastore):
    """Replace _DelayedUpload values with IncludedFile CAS handles
    (called once at run start, before the _parameters task persists)."""
    for name, value in list(values.items()):
        if not isinstance(value, _DelayedUpload):
            continue
        with open(value.path, "rb") as f:
            data = f.read()
        [(_uri, key)] = flow_datastore.save_data([data])
        values[name] = IncludedFile({
            "key": key,
            "size": len(data),
            "is_text": value.is_text,
            "encoding": value.encoding,
        })
    return values


class IncludeFile(Parameter):
    def __init__(self, name, required=False, is_text=True, encoding="utf-8",
                 default=None, help=None):
        super().__init__(name, default=default, required=required, help=help,
                         type=str)
        self.is_text = is_text
        self.encoding = encoding
        self.IS_INCLUDE_FILE = True

    def convert(self, value):
        if value is None:
            return None
        if isinstance(value, (IncludedFile, _DelayedUpload)):
            return value
        path = os.path.expanduser(str(value))
        if not os.path.isfile(path):
            from .exceptions import ParameterException

            raise ParameterException(
                "IncludeFile %s: no such file %r" % (self.name, path))
        return _DelayedUpload(path, self.is_text, self.encoding)
