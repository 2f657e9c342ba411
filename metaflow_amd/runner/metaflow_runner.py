"""Programmatic runner: drive a flow file's CLI from Python/notebooks.

Parity target: /root/reference/metaflow/runner/metaflow_runner.py:305
(Runner, ExecutingRun) and subprocess_manager.py — the runner writes a
`--runner-attribute-file`, launches the CLI as a subprocess, and wraps the
resulting run in the Client API.
"""

import json
import os
import subprocess
import sys
import tempfile
import time

from ..exceptions import MFXException


class ExecutingRun(object):
    def __init__(self, runner, process, attribute_file):
        self._runner = runner
        self.process = process
        self._attr_file = attribute_file
        self._attrs = None

    def _attributes(self, timeout=60):
        if self._attrs is None:
            deadline = time.time() + timeout
            while time.time() < deadline:
                try:
                    with open(self._attr_file) as f:
                        self._attrs = json.load(f)
                        break
                except (OSError, ValueError):
                    if self.process.poll() is not None:
                        break
                    time.sleep(0.1)
        return self._attrs or {}

    @property
    def run_id(self):
        return self._attributes().get("run_id")

    @property
    def run(self):
        """Client Run object for this execution."""
        from ..client import Run, namespace

        attrs = self._attributes()
        if not attrs:
            return None
        namespace(None)
        return Run("%s/%s" % (attrs["flow_name"], attrs["run_id"]))

    @property
    def returncode(self):
        return self.process.returncode

    @property
    def status(self):
        rc = self.process.poll()
        if rc is None:
            return "running"
        return "successful" if rc == 0 else "failed"

    @property
    def stdout(self):
        return self._out

    def wait(self, timeout=None):
        self._out, self._err = self.process.communicate(timeout=timeout)
        return self


class Runner(object):
    """Blocking programmatic runs:

        with Runner("flow.py", datastore_root=root) as r:
            result = r.run(alpha=5)
            print(result.run.data.final)
    """

    def __init__(self, flow_file, show_output=False, datastore_root=None,
                 env=None, cwd=None, **top_level_kwargs):
        self.flow_file = os.path.abspath(flow_file)
        if not os.path.exists(self.flow_file):
            raise MFXException("Flow file %s not found." % flow_file)
        self.show_output = show_output
        self.datastore_root = datastore_root
        self.env = env or {}
        self.cwd = cwd
        self.top_level_kwargs = top_level_kwargs

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        return False

    def _base_cmd(self):
        cmd = [sys.executable, self.flow_file]
        if self.datastore_root:
            cmd += ["--datastore-root", self.datastore_root]
        for k, v in self.top_level_kwargs.items():
            cmd += ["--%s" % k.replace("_", "-"), str(v)]
        return cmd

    def _launch(self, args):
        fd, attr_file = tempfile.mkstemp(suffix=".json")
        os.close(fd)
        os.unlink(attr_file)
        env = dict(os.environ)
        env.update({str(k): str(v) for k, v in self.env.items()})
        proc = subprocess.Popen(
            self._base_cmd() + args + ["--runner-attribute-file", attr_file],
            env=env,
            cwd=self.cwd,
            stdout=None if self.show_output else subprocess.PIPE,
            stderr=None if self.show_output else subprocess.PIPE,
            text=True,
        )
        return ExecutingRun(self, proc, attr_file)

    def run(self, **params):
        """Blocking run; returns the finished ExecutingRun."""
        ex = self.async_run(**params)
        ex.wait()
        if ex.returncode != 0:
            raise MFXException(
                "Run failed (rc=%d):\n%s" % (ex.returncode,
                                             (ex._err or "")[-2000:]))
        return ex

    def async_run(self, **params):
        args = ["run"]
        for k, v in params.items():
            args += ["--%s" % k.replace("_", "-"), str(v)]
        return self._launch(args)

    def resume(self, origin_run_id, **kwargs):
        args = ["resume", "--origin-run-id", str(origin_run_id)]
        for k, v in kwargs.items():
            args += ["--%s" % k.replace("_", "-"), str(v)]
        ex = self._launch(args)
        ex.wait()
        if ex.returncode != 0:
            raise MFXException("Resume failed (rc=%d):\n%s"
                               % (ex.returncode, (ex._err or "")[-2000:]))
        return ex
