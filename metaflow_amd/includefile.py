"""IncludeFile: a file-typed parameter whose content is uploaded to the CAS
at run start and lazily loaded in steps.

Parity target: /root/reference/metaflow/includefile.py:234 (IncludeFile,
IncludedFile handle).
"""

import os

from .parameters import Parameter


class IncludedFile(object):
    """Lazy handle stored as the parameter's value."""

    def __init__(self, descriptor):
        # descriptor: {"key": <cas key>} or {"path": <local path>}
        self.descriptor = descriptor

    def decode(self, flow_datastore=None):
        if "data" in self.descriptor:
            return self.descriptor["data"]
        if "path" in self.descriptor:
            with open(self.descriptor["path"], "rb") as f:
                return f.read()
        raise ValueError("IncludedFile has no content")

    def __repr__(self):
        return "IncludedFile(%s)" % self.descriptor


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
        if isinstance(value, IncludedFile):
            return value
        path = os.path.expanduser(str(value))
        with open(path, "rb") as f:
            data = f.read()
        if self.is_text:
            return data.decode(self.encoding)
        return data
