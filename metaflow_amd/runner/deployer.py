"""Deployer: register a flow as a named, triggerable deployment.

Parity target: /root/reference/metaflow/runner/deployer.py:99 (Deployer ->
DeployedFlow -> TriggeredRun). The reference's impls compile to external
orchestrators (Argo / Step Functions — out of scope per SURVEY §2.5); the
MI355X build backs the same three-object API with the LOCAL runtime: a
deployment is a content-addressed code package (package.CodePackage) plus
a deployment record under ``<ds_root>/_deployments/<flow>/<name>``, and a
trigger materializes that exact snapshot into a scratch dir and launches
it through Runner — so production runs are reproducible from the deployed
code, not from whatever the working tree has drifted to.

    Deployer("train_flow.py", datastore_root=root).local().create(
        name="prod")                       # -> DeployedFlow
    df = DeployedFlow.get("TrainFlow", "prod", datastore_root=root)
    tr = df.trigger(alpha=5)               # -> TriggeredRun (async)
    tr.wait(); print(tr.run.data.result)
"""

import json
import os
import shutil
import sys
import tempfile
import time

from ..exceptions import MFXException


def _deploy_dir(datastore_root, flow_name, name):
    return os.path.join(datastore_root, "_deployments", flow_name, name)


def _flow_name_of(flow_file):
    """The FlowSpec subclass name, found the same way the CLI does: the
    last FlowSpec subclass defined at the flow file's top level."""
    import ast

    with open(flow_file) as f:
        tree = ast.parse(f.read())
    name = None
    for node in tree.body:
        if isinstance(node, ast.ClassDef):
            for base in node.bases:
                base_name = getattr(base, "id", getattr(base, "attr", ""))
                if base_name == "FlowSpec" or base_name.endswith("FlowSpec"):
                    name = node.name
    if name is None:
        raise MFXException("No FlowSpec subclass found in %s" % flow_file)
    return name


class TriggeredRun(object):
    """Handle on one triggered execution of a deployed flow (async: the
    subprocess is already running when trigger() returns)."""

    def __init__(self, deployed_flow, executing_run, workdir):
        self.deployed_flow = deployed_flow
        self._ex = executing_run
        self._workdir = workdir

    @property
    def run_id(self):
        return self._ex.run_id

    @property
    def status(self):
        return self._ex.status

    @property
    def run(self):
        return self._ex.run

    def wait(self, timeout=None):
        self._ex.wait(timeout=timeout)
        if self._workdir:
            shutil.rmtree(self._workdir, ignore_errors=True)
            self._workdir = None
        if self._ex.returncode != 0:
            raise MFXException(
                "Triggered run failed (rc=%d):\n%s"
                % (self._ex.returncode, (self._ex._err or "")[-2000:]))
        return self


class DeployedFlow(object):
    def __init__(self, flow_name, name, record, datastore_root):
        self.flow_name = flow_name
        self.name = name
        self._record = record
        self._datastore_root = datastore_root

    @property
    def code_package_key(self):
        return self._record["code_package_key"]

    @property
    def created_at(self):
        return self._record["created_at"]

    @classmethod
    def get(cls, flow_name, name="prod", datastore_root=None):
        datastore_root = datastore_root or os.environ.get(
            "MFX_DATASTORE_ROOT", ".mfx")
        path = os.path.join(_deploy_dir(datastore_root, flow_name, name),
                            "deployment.json")
        try:
            with open(path) as f:
                record = json.load(f)
        except OSError:
            raise MFXException("No deployment %s/%s under %s"
                               % (flow_name, name, datastore_root))
        return cls(flow_name, name, record, datastore_root)

    @classmethod
    def list_deployed(cls, flow_name, datastore_root=None):
        datastore_root = datastore_root or os.environ.get(
            "MFX_DATASTORE_ROOT", ".mfx")
        base = os.path.join(datastore_root, "_deployments", flow_name)
        try:
            names = sorted(os.listdir(base))
        except OSError:
            return []
        return [cls.get(flow_name, n, datastore_root) for n in names
                if os.path.isfile(os.path.join(base, n,
                                               "deployment.json"))]

    def _materialize(self):
        """Extract the deployed code snapshot into a scratch dir."""
        from ..datastore.flow_datastore import FlowDataStore
        from ..package import CodePackage
        from ..datastore.storage import LocalStorage

        fds = FlowDataStore(self.flow_name,
                            LocalStorage(self._datastore_root))
        dest = tempfile.mkdtemp(prefix="mfx_deploy_")
        CodePackage.extract(fds, self.code_package_key, dest)
        return dest

    def trigger(self, **params):
        """Launch one run of the DEPLOYED code snapshot (async);
        returns a TriggeredRun."""
        from .metaflow_runner import Runner

        workdir = self._materialize()
        flow_file = os.path.join(workdir, self._record["flow_file"])
        runner = Runner(flow_file, datastore_root=self._datastore_root,
                        cwd=workdir,
                        env=self._record.get("env") or {})
        ex = runner.async_run(**params)
        return TriggeredRun(self, ex, workdir)

    def delete(self):
        shutil.rmtree(_deploy_dir(self._datastore_root, self.flow_name,
                                  self.name), ignore_errors=True)


class _LocalDeployerImpl(object):
    def __init__(self, deployer):
        self._d = deployer

    def create(self, name="prod"):
        """Package the flow dir, store it in the CAS, write the
        deployment record; returns a DeployedFlow."""
        from ..datastore.flow_datastore import FlowDataStore
        from ..package import CodePackage
        from ..datastore.storage import LocalStorage

        d = self._d
        flow_name = _flow_name_of(d.flow_file)
        fds = FlowDataStore(flow_name, LocalStorage(d.datastore_root))
        pkg = CodePackage(os.path.dirname(d.flow_file))
        _uri, key = pkg.save(fds)
        record = {
            "flow_name": flow_name,
            "name": name,
            "flow_file": os.path.basename(d.flow_file),
            "code_package_key": key,
            "created_at": time.strftime("%Y-%m-%dT%H:%M:%S"),
            "env": d.env,
            "python": sys.version.split()[0],
        }
        dest = _deploy_dir(d.datastore_root, flow_name, name)
        os.makedirs(dest, exist_ok=True)
        tmp = os.path.join(dest, ".deployment.json.tmp")
        with open(tmp, "w") as f:
            json.dump(record, f, indent=1)
        os.replace(tmp, os.path.join(dest, "deployment.json"))
        return DeployedFlow(flow_name, name, record, d.datastore_root)


class Deployer(object):
    """Entry point mirroring the reference's impl-per-backend shape:
    ``Deployer(file).local()`` returns the local-runtime backend (the
    only one in the MI355X build — orchestrator compilers are out of
    scope)."""

    def __init__(self, flow_file, datastore_root=None, env=None,
                 **top_level_kwargs):
        self.flow_file = os.path.abspath(flow_file)
        if not os.path.exists(self.flow_file):
            raise MFXException("Flow file %s not found." % flow_file)
        self.datastore_root = datastore_root or os.environ.get(
            "MFX_DATASTORE_ROOT", ".mfx")
        self.env = env or {}
        self.top_level_kwargs = top_level_kwargs

    def local(self):
        return _LocalDeployerImpl(self)
