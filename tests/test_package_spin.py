"""Code packaging, spin, and tag CLI tests."""

import os

from .test_runtime import latest_run_id, read_artifact, run_flow


def test_code_package_roundtrip(tmp_datastore, tmp_path):
    run_flow("linear_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    run = client.Run("LinearFlow/%s" % run_id)
    key = run.code_package_key
    assert key
    dest = str(tmp_path / "code")
    run.extract_code(dest)
    assert os.path.exists(os.path.join(dest, "linear_flow.py"))
    assert os.path.exists(os.path.join(dest, "MFX_MANIFEST.json"))


def test_package_deterministic(tmp_path):
    from metaflow_amd.package import CodePackage

    d = tmp_path / "src"
    d.mkdir()
    (d / "a.py").write_text("print(1)\n")
    (d / "b.py").write_text("print(2)\n")
    b1 = CodePackage(str(d)).blob()
    b2 = CodePackage(str(d)).blob()
    assert b1 == b2


def test_spin(tmp_datastore):
    run_flow("linear_flow.py", tmp_datastore, "run", "--alpha", "4")
    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    proc = run_flow("linear_flow.py", tmp_datastore, "spin", "middle",
                    "--run-id", run_id)
    assert "spin task done" in proc.stdout
    assert read_artifact(tmp_datastore, "LinearFlow", "spin%s" % run_id,
                         "middle", "x") == 40


def test_tag_cli(tmp_datastore):
    run_flow("linear_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    run_flow("linear_flow.py", tmp_datastore, "tag", "add", run_id,
             "experiment:x1")
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    run = client.Run("LinearFlow/%s" % run_id)
    assert "experiment:x1" in run.tags
    run.remove_tags(["experiment:x1"])
    assert "experiment:x1" not in client.Run("LinearFlow/%s" % run_id).tags
    # singular + atomic replace (reference Run.add_tag/replace_tag)
    run.add_tag("phase:a")
    assert "phase:a" in client.Run("LinearFlow/%s" % run_id).tags
    run.replace_tag("phase:a", "phase:b")
    tags = client.Run("LinearFlow/%s" % run_id).tags
    assert "phase:b" in tags and "phase:a" not in tags
    run.remove_tag("phase:b")
    assert "phase:b" not in client.Run("LinearFlow/%s" % run_id).tags


def test_spin_foreach_task(tmp_datastore):
    """spin a single foreach child against origin artifacts via
    --split-index."""
    run_flow("foreach_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "ForeachFlow")
    proc = run_flow("foreach_flow.py", tmp_datastore, "spin", "work",
                    "--run-id", run_id, "--split-index", "2")
    assert proc.returncode == 0
    assert read_artifact(tmp_datastore, "ForeachFlow",
                         "spin%s" % run_id, "work", "squared") == 9
