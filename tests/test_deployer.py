"""Deployer -> DeployedFlow -> TriggeredRun (reference
runner/deployer.py:99 pattern, backed by the local runtime): a
deployment is a content-addressed code snapshot; triggers run THAT
snapshot even after the working tree drifts."""

import os
import textwrap

import pytest

from .test_runtime import REPO


@pytest.fixture()
def flow_dir(tmp_path):
    d = tmp_path / "proj"
    d.mkdir()
    (d / "dep_flow.py").write_text(textwrap.dedent("""
        from metaflow_amd import FlowSpec, Parameter, step

        class DepFlow(FlowSpec):
            alpha = Parameter("alpha", default=2, type=int)

            @step
            def start(self):
                self.out = self.alpha * 10
                self.version = "v1"
                self.next(self.end)

            @step
            def end(self):
                pass

        if __name__ == "__main__":
            DepFlow()
    """))
    return d


def _env(monkeypatch, tmp_datastore):
    monkeypatch.setenv("PYTHONPATH",
                       REPO + os.pathsep + os.environ.get("PYTHONPATH", ""))
    monkeypatch.setenv("MFX_NUM_GPUS", "0")
    monkeypatch.setenv("MFX_DATASTORE_ROOT", tmp_datastore)


def test_deploy_trigger_wait(flow_dir, tmp_datastore, monkeypatch):
    from metaflow_amd import DeployedFlow, Deployer

    _env(monkeypatch, tmp_datastore)
    df = Deployer(str(flow_dir / "dep_flow.py"),
                  datastore_root=tmp_datastore).local().create(name="prod")
    assert df.flow_name == "DepFlow" and df.code_package_key

    got = DeployedFlow.get("DepFlow", "prod", datastore_root=tmp_datastore)
    assert got.code_package_key == df.code_package_key
    assert [d.name for d in
            DeployedFlow.list_deployed("DepFlow",
                                       datastore_root=tmp_datastore)] == \
        ["prod"]

    tr = got.trigger(alpha=7)
    tr.wait(timeout=300)
    assert tr.status == "successful"
    assert tr.run["start"].task.data.out == 70


def test_trigger_runs_deployed_snapshot_not_workdir(flow_dir,
                                                    tmp_datastore,
                                                    monkeypatch):
    """Edit the flow file AFTER deploying: the trigger still runs the
    deployed v1 code (reproducible production runs)."""
    from metaflow_amd import Deployer

    _env(monkeypatch, tmp_datastore)
    df = Deployer(str(flow_dir / "dep_flow.py"),
                  datastore_root=tmp_datastore).local().create(name="prod")
    # drift the working tree
    src = (flow_dir / "dep_flow.py").read_text()
    (flow_dir / "dep_flow.py").write_text(src.replace('"v1"', '"v2"'))
    tr = df.trigger()
    tr.wait(timeout=300)
    assert tr.run["start"].task.data.version == "v1"


def test_get_missing_deployment_raises(tmp_datastore, monkeypatch):
    from metaflow_amd import DeployedFlow
    from metaflow_amd.exceptions import MFXException

    _env(monkeypatch, tmp_datastore)
    with pytest.raises(MFXException):
        DeployedFlow.get("NoSuchFlow", "prod",
                         datastore_root=tmp_datastore)


def test_mfx_deploy_cli(flow_dir, tmp_datastore, monkeypatch):
    """mfx deploy / deployments / trigger round-trip."""
    import subprocess
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    env["MFX_DATASTORE_ROOT"] = tmp_datastore

    def mfx(*args):
        return subprocess.run(
            [sys.executable, "-m", "metaflow_amd.cmd"] + list(args),
            env=env, capture_output=True, text=True, timeout=300)

    r = mfx("deploy", str(flow_dir / "dep_flow.py"), "--name", "prod")
    assert r.returncode == 0 and "deployed DepFlow/prod" in r.stdout, \
        r.stderr[-2000:]
    r = mfx("deployments", "DepFlow")
    assert r.returncode == 0 and "prod" in r.stdout
    r = mfx("trigger", "DepFlow", "--param", "alpha=3")
    assert r.returncode == 0 and "successful" in r.stdout, r.stderr[-2000:]


def test_nbrunner_and_nbdeployer(tmp_datastore, monkeypatch):
    """NBRunner/NBDeployer: a class defined in THIS file (stand-in for
    a notebook cell) runs and deploys without a user-written flow
    file."""
    from metaflow_amd import NBDeployer, NBRunner

    _env(monkeypatch, tmp_datastore)

    from metaflow_amd import FlowSpec, Parameter, step

    class CellFlow(FlowSpec):
        alpha = Parameter("alpha", default=2, type=int)

        @step
        def start(self):
            self.out = self.alpha * 11
            self.next(self.end)

        @step
        def end(self):
            pass

    run = NBRunner(CellFlow, datastore_root=tmp_datastore).nbrun(alpha=4)
    assert run["start"].task.data.out == 44

    df = NBDeployer(CellFlow,
                    datastore_root=tmp_datastore).deploy(name="nb")
    tr = df.trigger(alpha=6)
    tr.wait(timeout=300)
    assert tr.run["start"].task.data.out == 66
