"""Code packaging: a content-addressed tarball of the flow's code.

Parity target: /root/reference/metaflow/package/__init__.py:43
(MetaflowPackage). The package is stored through FlowDataStore.save_data
(raw CAS blob) and recorded in run metadata, so any run can be reproduced
from its exact code snapshot.
"""

import io
import json
import os
import tarfile

DEFAULT_SUFFIXES = (".py", ".txt", ".yaml", ".yml", ".json", ".hip", ".cpp",
                    ".h", ".sh", ".toml", ".cfg")
EXCLUDE_DIRS = {".git", "__pycache__", ".mfx", "gpurun_out", "build",
                ".pytest_cache"}


def _walk(root, suffixes):
    for dirpath, dirnames, filenames in os.walk(root, followlinks=False):
        dirnames[:] = [d for d in dirnames if d not in EXCLUDE_DIRS
                       and not d.startswith(".")]
        for name in filenames:
            if name.endswith(suffixes):
                full = os.path.join(dirpath, name)
                rel = os.path.relpath(full, root)
                yield full, rel


class CodePackage(object):
    def __init__(self, flow_dir, suffixes=DEFAULT_SUFFIXES,
                 max_file_size=8 << 20):
        self.flow_dir = os.path.abspath(flow_dir)
        self.suffixes = tuple(suffixes)
        self.max_file_size = max_file_size
        self._blob = None

    def blob(self):
        """Deterministic tar (sorted names, zeroed mtimes) so identical
        code -> identical CAS key."""
        if self._blob is None:
            buf = io.BytesIO()
            with tarfile.open(fileobj=buf, mode="w:gz",
                              compresslevel=3) as tar:
                entries = sorted(_walk(self.flow_dir, self.suffixes),
                                 key=lambda x: x[1])
                manifest = {"files": [rel for _f, rel in entries],
                            "created": 0}
                info = tarfile.TarInfo("MFX_MANIFEST.json")
                data = json.dumps(manifest).encode()
                info.size = len(data)
                tar.addfile(info, io.BytesIO(data))
                for full, rel in entries:
                    if os.path.getsize(full) > self.max_file_size:
                        continue
                    info = tar.gettarinfo(full, arcname=rel)
                    info.mtime = 0
                    info.uid = info.gid = 0
                    info.uname = info.gname = ""
                    with open(full, "rb") as f:
                        tar.addfile(info, f)
            self._blob = buf.getvalue()
        return self._blob

    def save(self, flow_datastore):
        """Store in the CAS; returns (uri, key)."""
        [(uri, key)] = flow_datastore.save_data([self.blob()])
        return uri, key

    @staticmethod
    def extract(flow_datastore, key, dest):
        [(_k, blob)] = flow_datastore.load_data([key])
        os.makedirs(dest, exist_ok=True)
        with tarfile.open(fileobj=io.BytesIO(blob), mode="r:gz") as tar:
            tar.extractall(dest)
        return dest
