"""Nested foreach, switch, Runner API, client API, and the CPU gang train
flow (gloo world=2)."""

import os

from .test_runtime import FLOWS, REPO, latest_run_id, read_artifact, run_flow


def test_nested_foreach(tmp_datastore):
    run_flow("nested_foreach_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "NestedForeachFlow")
    assert read_artifact(tmp_datastore, "NestedForeachFlow", run_id,
                         "outer_join", "total") == 180


def test_switch_fast(tmp_datastore):
    run_flow("switch_flow.py", tmp_datastore, "run", "--route", "fast")
    run_id = latest_run_id(tmp_datastore, "SwitchFlow")
    assert read_artifact(tmp_datastore, "SwitchFlow", run_id, "finish",
                         "final") == "fast"


def test_switch_slow(tmp_datastore):
    run_flow("switch_flow.py", tmp_datastore, "run", "--route", "slow")
    run_id = latest_run_id(tmp_datastore, "SwitchFlow")
    assert read_artifact(tmp_datastore, "SwitchFlow", run_id, "finish",
                         "final") == "slow"


def test_runner_api(tmp_datastore, monkeypatch):
    monkeypatch.setenv("MFX_NUM_GPUS", "0")
    monkeypatch.setenv(
        "PYTHONPATH", REPO + os.pathsep + os.environ.get("PYTHONPATH", ""))
    from metaflow_amd.runner import Runner

    r = Runner(os.path.join(FLOWS, "linear_flow.py"),
               datastore_root=tmp_datastore)
    ex = r.run(alpha=7)
    assert ex.status == "successful"
    run = ex.run
    assert run is not None
    task = run["end"].task
    assert task.data.final == 71
    assert task.successful


def test_client_api(tmp_datastore):
    run_flow("branch_flow.py", tmp_datastore, "run")
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)

    mf = client.Metaflow()
    flows = [f.id for f in mf]
    assert "BranchFlow" in flows
    # the persisted DAG travels with the run (reference runtime_dag)
    dag = client.Flow("BranchFlow").latest_run.dag
    assert dag and dag["start"]["out_funcs"] == ["a", "b"]
    assert dag["join"]["type"] == "join"
    flow = client.Flow("BranchFlow")
    run = flow.latest_run
    assert run.successful
    steps = {s.id for s in run}
    assert {"start", "a", "b", "join", "end"} <= steps
    join_task = run["join"].task
    assert join_task.data.total == 3
    art = join_task["total"]
    assert art.data == 3 and art.sha and art.size > 0
    # namespace filtering
    client.namespace("user:nonexistent-user")
    try:
        client.Flow("BranchFlow").latest_run
        got = client.Flow("BranchFlow").latest_run
        assert got is None
    finally:
        client.namespace(None)


def test_gang_train_flow_cpu(tmp_datastore):
    """The full @parallel + @torch_parallel + @checkpoint path on CPU:
    gang of 2 over gloo through the real scheduler."""
    run_flow("train_flow.py", tmp_datastore, "run", "--num-nodes", "2",
             "--steps-n", "2", timeout=420)
    run_id = latest_run_id(tmp_datastore, "TrainFlow")
    all_losses = read_artifact(tmp_datastore, "TrainFlow", run_id, "join",
                               "all_losses")
    assert set(all_losses) == {0, 1}
    # checkpoint artifact exists on the control task
    idx = read_artifact(tmp_datastore, "TrainFlow", run_id, "train",
                        "_checkpoint_final_index")
    assert "flat_param" in idx and "adam_m" in idx


def test_unbounded_foreach(tmp_datastore):
    run_flow("ubf_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "UBFFlow")
    assert read_artifact(tmp_datastore, "UBFFlow", run_id, "join",
                         "total") == 30
    # control + 3 mappers exist under the work step
    work_dir = os.path.join(tmp_datastore, "UBFFlow", run_id, "work")
    tasks = os.listdir(work_dir)
    assert len([t for t in tasks if "_mapper_" in t]) == 3
    ctl = read_artifact(tmp_datastore, "UBFFlow", run_id, "work",
                        "_control_mapper_tasks")
    assert len(ctl) == 3


def test_card(tmp_datastore):
    from metaflow_amd.datastore import FlowDataStore
    from metaflow_amd.datastore.storage import LocalStorage
    from metaflow_amd.plugins.card_decorator import get_card

    run_flow("card_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "CardFlow")
    fds = FlowDataStore("CardFlow", LocalStorage(tmp_datastore))
    task_id = fds.list_tasks(run_id, "start")[0]
    ds = fds.get_task_datastore(run_id, "start", task_id)
    html = get_card(ds)
    assert html and "metric" in html and "custom html" in html
    assert "Notes" in html and "OK" in html
    # typed components (card_components.py)
    assert "<h3>Results</h3>" in html          # Markdown heading
    assert "<b>0.125</b>" in html              # Markdown bold
    assert "<th>a</th>" in html and "<td>4</td>" in html   # Table
    assert "data:image/png;base64," in html    # Image data URI
    assert "config" in html and "3e-04" in html.replace(
        "0.0003", "3e-04")                     # Artifact pprint
    # a second card with an explicit id on another step
    end_task = fds.list_tasks(run_id, "end")[0]
    ds_end = fds.get_task_datastore(run_id, "end", end_task)
    html2 = get_card(ds_end, card_id="report")
    assert html2 and "Final report" in html2


def test_config_and_mutator(tmp_datastore, tmp_path):
    run_flow("config_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "ConfigFlow")
    assert read_artifact(tmp_datastore, "ConfigFlow", run_id, "end",
                         "value") == 30
    assert read_artifact(tmp_datastore, "ConfigFlow", run_id, "end",
                         "retry_attached") is True
    # config file override via --config
    cfgf = tmp_path / "override.json"
    cfgf.write_text('{"scale": 7, "retries": 0}')
    run_flow("config_flow.py", tmp_datastore, "--config",
             "cfg=%s" % cfgf, "run")
    run_id = latest_run_id(tmp_datastore, "ConfigFlow")
    assert read_artifact(tmp_datastore, "ConfigFlow", run_id, "end",
                         "value") == 70


def test_client_lineage_and_liveness(tmp_datastore):
    run_flow("foreach_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "ForeachFlow")
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    run = client.Run("ForeachFlow/%s" % run_id)
    assert not run.is_alive  # finished
    join_task = run["join"].task
    parents = join_task.parent_tasks
    assert len(parents) == 4  # the four foreach work tasks
    work0 = run["work"].tasks()[0]
    kids = work0.child_tasks
    assert any(t.step_name == "join" for t in kids)


def test_card_server(tmp_datastore):
    import subprocess
    import sys
    import time
    import urllib.request

    run_flow("card_flow.py", tmp_datastore, "run")
    import socket

    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    proc = subprocess.Popen(
        [sys.executable, "-m", "metaflow_amd", "card-server",
         "--port", str(port)], env=env,
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    try:
        body = None
        for _ in range(100):
            try:
                body = urllib.request.urlopen(
                    "http://127.0.0.1:%d/" % port, timeout=1).read()
                break
            except Exception:
                time.sleep(0.1)
        assert body and b"CardFlow" in body
        link = body.decode().split('href="/')[1].split('"')[0]
        card = urllib.request.urlopen(
            "http://127.0.0.1:%d/%s" % (port, link), timeout=5).read()
        assert b"custom html" in card
    finally:
        proc.terminate()
        proc.wait(timeout=5)


def test_torch_parallel_cp_grid(tmp_datastore):
    """@torch_parallel(context_parallel=2): dp x cp grid groups exposed on
    current.parallel, collective + ring attention run over cp_group."""
    run_flow("cp_grid_flow.py", tmp_datastore, "run", timeout=300)


def test_example_train_llama_cp(tmp_datastore):
    """The long-context example flow end-to-end on CPU (2-rank ring)."""
    import subprocess
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "train_llama_cp.py"),
         "--quiet", "--datastore-root", tmp_datastore, "run",
         "--num-gpus", "2", "--train-steps", "2", "--seq", "256"],
        capture_output=True, text=True, env=env, timeout=300)
    assert proc.returncode == 0, proc.stderr[-2000:]
    run_id = latest_run_id(tmp_datastore, "TrainLlamaCP")
    losses = read_artifact(tmp_datastore, "TrainLlamaCP", run_id, "join",
                           "losses")
    assert len(losses) == 2 and all(l == l for l in losses)


def test_recursive_switch(tmp_datastore):
    """A switch back-edge loops a step until its condition flips
    (reference recursive_switch): three iterations, then the exit arm."""
    run_flow("recursive_switch_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "RecursiveSwitchFlow")
    assert read_artifact(tmp_datastore, "RecursiveSwitchFlow", run_id,
                         "finish", "total") == 30
    # three work iterations really ran as three tasks
    work_dir = os.path.join(tmp_datastore, "RecursiveSwitchFlow", run_id,
                            "work")
    assert len(os.listdir(work_dir)) == 3


def test_recursive_switch_resume(tmp_datastore):
    """Resume of a flow containing a switch loop: the successful loop
    iterations clone, only the failed tail reruns."""
    proc = run_flow("recursive_switch_flow.py", tmp_datastore, "run",
                    check=False, env_extra={"REC_FAIL": "1"})
    assert proc.returncode != 0
    proc2 = run_flow("recursive_switch_flow.py", tmp_datastore, "resume",
                     env_extra={"REC_FAIL": "0"})
    assert proc2.returncode == 0
    run_id = latest_run_id(tmp_datastore, "RecursiveSwitchFlow")
    assert read_artifact(tmp_datastore, "RecursiveSwitchFlow", run_id,
                         "finish", "total") == 30


def test_project_namespacing(tmp_datastore):
    """@project: current.project_* populated in tasks and the run tagged
    with project:/project_branch: (reference project_branch behavior)."""
    run_flow("project_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "ProjectFlow")
    assert read_artifact(tmp_datastore, "ProjectFlow", run_id, "start",
                         "pname") == "mlplat"
    branch = read_artifact(tmp_datastore, "ProjectFlow", run_id, "start",
                           "branch")
    assert branch.startswith("user.")
    assert read_artifact(tmp_datastore, "ProjectFlow", run_id, "start",
                         "pflow") == "mlplat.%s.ProjectFlow" % branch
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    tags = client.Run("ProjectFlow/%s" % run_id).tags
    assert "project:mlplat" in tags
    assert any(t.startswith("project_branch:user.") for t in tags)


def test_client_task_logs(tmp_datastore, tmp_path):
    """Client Task.stdout/loglines return the captured user prints
    (reference basic_log behavior)."""
    import subprocess
    import sys

    flow = tmp_path / "logflow.py"
    flow.write_text(
        "import sys\n"
        "from metaflow_amd import FlowSpec, step\n"
        "class LogFlow(FlowSpec):\n"
        "    @step\n"
        "    def start(self):\n"
        "        print('hello-from-start')\n"
        "        sys.stderr.write('warn-from-start\\n')\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def end(self):\n"
        "        pass\n"
        "if __name__ == '__main__':\n"
        "    LogFlow()\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        capture_output=True, text=True, env=env, timeout=180)
    assert proc.returncode == 0, proc.stderr[-1500:]
    run_id = latest_run_id(tmp_datastore, "LogFlow")
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    task = client.Run("LogFlow/%s" % run_id)["start"].task
    assert "hello-from-start" in task.stdout
    assert "warn-from-start" in task.stderr
    lines = list(task.loglines("stdout"))
    assert any("hello-from-start" in str(l) for l in lines)


def test_nested_unbounded_foreach(tmp_datastore):
    """UBF control/mapper fan-out nested inside a static foreach: per-
    branch mappers, two-level join (reference nested_unbounded_foreach)."""
    run_flow("nested_ubf_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "NestedUBFFlow")
    assert read_artifact(tmp_datastore, "NestedUBFFlow", run_id, "join_o",
                         "total") == 96
    # two control tasks (one per outer branch), each with 3 mappers
    work_dir = os.path.join(tmp_datastore, "NestedUBFFlow", run_id, "work")
    tasks = os.listdir(work_dir)
    assert len([t for t in tasks if "_mapper_" in t]) == 6


def test_example_train_llama_ddp(tmp_datastore):
    """The DDP example flow end-to-end on CPU (2-rank gloo gang)."""
    import subprocess
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable,
         os.path.join(REPO, "examples", "train_llama_ddp.py"),
         "--quiet", "--datastore-root", tmp_datastore, "run",
         "--num-gpus", "2", "--train-steps", "2", "--batch", "1",
         "--seq", "256", "--model-size", "tiny"],
        capture_output=True, text=True, env=env, timeout=420)
    assert proc.returncode == 0, proc.stderr[-2000:]


def test_gang_gpu_pinning(tmp_datastore):
    """@resources(gpu=2) on a 2-rank gang with 4 (simulated) GPUs pins
    disjoint device pairs per rank."""
    run_flow("gpu_pin_flow.py", tmp_datastore, "run",
             env_extra={"MFX_NUM_GPUS": "4"})


def test_rocprof_stats_section(tmp_path):
    """Kernel-stats CSV -> HTML breakdown table (the @card(profile=True)
    post-exit splice)."""
    from metaflow_amd.plugins.card_decorator import rocprof_stats_section

    d = tmp_path / "prof"
    d.mkdir()
    (d / "run_kernel_stats.csv").write_text(
        '"Name","Calls","TotalDurationNs","AverageNs","Percentage",'
        '"MinNs","MaxNs","StdDev"\n'
        '"attn_fwd_v2_kernel",96,500000000,5208333,41.7,1,2,0.1\n'
        '"Cijk_gemm_big",291,400000000,1374570,33.3,1,2,0.1\n'
        '"tiny_kernel",10,1000,100,0.0,1,2,0.0\n')
    html = rocprof_stats_section(str(d))
    assert html is not None
    assert "attn_fwd_v2_kernel" in html
    assert html.index("attn_fwd_v2_kernel") < html.index("Cijk_gemm_big")
    assert "500.0" in html  # total ms


def test_card_profile_splice(tmp_path):
    """get_card splices the scheduler-saved rocprof section into the
    rendered card."""
    from metaflow_amd.datastore import FlowDataStore
    from metaflow_amd.datastore.storage import LocalStorage
    from metaflow_amd.plugins.card_decorator import get_card

    fds = FlowDataStore("CardFlow", LocalStorage(str(tmp_path)))
    ds = fds.get_task_datastore("1", "s", "1", attempt=0, mode="w")
    ds.init_task()
    ds.save_metadata("card_default",
                     {"html": "<html><body><h1>t</h1></body></html>"})
    ds.save_metadata("card_profile", {"html": "<table>KERNELS</table>"})
    ds.done()
    rd = fds.get_task_datastore("1", "s", "1")
    doc = get_card(rd)
    assert "KERNELS" in doc
    assert "Kernel-time breakdown" in doc


def test_client_card_html(tmp_datastore):
    """Client Task.card_html returns the rendered card."""
    import os

    from .test_runtime import latest_run_id, run_flow

    run_flow("card_flow.py", tmp_datastore, "run")
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    try:
        from metaflow_amd.client import Flow, namespace

        namespace(None)
        run_id = latest_run_id(tmp_datastore, "CardFlow")
        flow = Flow("CardFlow")
        task = flow[run_id]["start"].task
        doc = task.card_html
        assert doc and "<html" in doc
    finally:
        del os.environ["MFX_DATASTORE_SYSROOT_LOCAL"]
