"""Notebook entry points: run or deploy a FlowSpec CLASS defined in the
current process (a notebook cell or REPL) without a flow file on disk.

Parity target: /root/reference/metaflow/runner/nbrun.py (NBRunner) and
nbdeploy.py (NBDeployer): extract the class source, materialize a
self-contained flow file in a scratch dir, and drive the file-based
Runner/Deployer. Same contract as the reference: the class body must be
self-contained — imports your steps need should happen inside the steps
(the generated file imports only the metaflow_amd public API).
"""

import inspect
import os
import shutil
import tempfile
import textwrap

from ..exceptions import MFXException

_PRELUDE = "from metaflow_amd import *  # noqa: F401,F403\n\n\n"


def _materialize(flow_cls, base_dir=None):
    try:
        source = inspect.getsource(flow_cls)
    except (OSError, TypeError) as ex:
        raise MFXException(
            "Cannot extract source for %r — NBRunner needs the class "
            "defined in a cell/file whose source is available (%s)"
            % (flow_cls, ex))
    body = _PRELUDE + textwrap.dedent(source) + (
        "\n\nif __name__ == '__main__':\n    %s()\n" % flow_cls.__name__)
    workdir = tempfile.mkdtemp(prefix="mfx_nb_", dir=base_dir)
    path = os.path.join(workdir, "notebook_flow.py")
    with open(path, "w") as f:
        f.write(body)
    return workdir, path


class NBRunner(object):
    """Run a notebook-defined flow:

        class MyFlow(FlowSpec): ...
        run = NBRunner(MyFlow, datastore_root=root).nbrun(alpha=3)
        run.data  # Client Run object
    """

    def __init__(self, flow_cls, datastore_root=None, env=None,
                 base_dir=None, **top_level_kwargs):
        from .metaflow_runner import Runner

        self._workdir, path = _materialize(flow_cls, base_dir)
        self.runner = Runner(path, datastore_root=datastore_root,
                             env=env, **top_level_kwargs)

    def nbrun(self, **params):
        """Blocking run; returns the Client Run object."""
        try:
            return self.runner.run(**params).run
        finally:
            self.cleanup()

    def async_run(self, **params):
        """Returns the ExecutingRun (caller must cleanup())."""
        return self.runner.async_run(**params)

    def cleanup(self):
        if self._workdir:
            shutil.rmtree(self._workdir, ignore_errors=True)
            self._workdir = None


class NBDeployer(object):
    """Deploy a notebook-defined flow through the local Deployer:

        df = NBDeployer(MyFlow, datastore_root=root).deploy(name="prod")
        df.trigger(...)
    """

    def __init__(self, flow_cls, datastore_root=None, env=None,
                 base_dir=None, **kwargs):
        from .deployer import Deployer

        self._workdir, path = _materialize(flow_cls, base_dir)
        self.deployer = Deployer(path, datastore_root=datastore_root,
                                 env=env, **kwargs)

    def deploy(self, name="prod"):
        """Create the deployment (code snapshot of the materialized
        flow); returns the DeployedFlow."""
        try:
            return self.deployer.local().create(name=name)
        finally:
            self.cleanup()

    def cleanup(self):
        if self._workdir:
            shutil.rmtree(self._workdir, ignore_errors=True)
            self._workdir = None
