"""Retry, catch, and resume recovery tests (CPU, subprocess-driven)."""

import os

from .test_runtime import REPO, latest_run_id, read_artifact, run_flow


def test_retry(tmp_datastore, tmp_path, monkeypatch):
    marker_dir = tmp_path / "markers"
    marker_dir.mkdir()
    monkeypatch.setenv("RETRY_MARKER_DIR", str(marker_dir))
    run_flow("retry_flow.py", tmp_datastore, "run")
    attempts = sorted(os.listdir(str(marker_dir)))
    assert attempts == ["attempt_0", "attempt_1", "attempt_2"]
    run_id = latest_run_id(tmp_datastore, "RetryFlow")
    assert read_artifact(tmp_datastore, "RetryFlow", run_id, "flaky",
                         "attempts_seen") == 2


def test_catch(tmp_datastore):
    run_flow("catch_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "CatchFlow")
    failure = read_artifact(tmp_datastore, "CatchFlow", run_id, "will_fail",
                            "failure")
    assert "intentional" in failure.exception


def test_resume(tmp_datastore, tmp_path, monkeypatch):
    counter_dir = tmp_path / "counters"
    counter_dir.mkdir()
    monkeypatch.setenv("RESUME_COUNTER_DIR", str(counter_dir))

    monkeypatch.setenv("RESUME_FAIL", "1")
    proc = run_flow("resume_flow.py", tmp_datastore, "run", check=False)
    assert proc.returncode != 0
    orig_run = latest_run_id(tmp_datastore, "ResumeFlow")

    monkeypatch.setenv("RESUME_FAIL", "0")
    run_flow("resume_flow.py", tmp_datastore, "resume",
             "--origin-run-id", orig_run)
    new_run = latest_run_id(tmp_datastore, "ResumeFlow")
    assert new_run != orig_run

    # start was cloned (ran once), middle ran twice, end once
    assert int(open(counter_dir / "start").read()) == 1
    assert int(open(counter_dir / "middle").read()) == 2
    assert int(open(counter_dir / "end").read()) == 1
    assert read_artifact(tmp_datastore, "ResumeFlow", new_run, "end",
                         "z") == 11


def test_with_retry_attach(tmp_datastore, tmp_path, monkeypatch):
    """--with retry attaches the decorator to all steps."""
    proc = run_flow("linear_flow.py", tmp_datastore,
                    "--with", "retry:times=1", "run")
    assert proc.returncode == 0


def test_reentrant_resume(tmp_datastore, tmp_path, monkeypatch):
    """Two concurrent `resume --reentrant` calls: exactly one executes,
    the other waits and exits cleanly; tasks ran once."""
    import subprocess
    import sys

    counter_dir = tmp_path / "counters"
    counter_dir.mkdir()
    monkeypatch.setenv("RESUME_COUNTER_DIR", str(counter_dir))
    monkeypatch.setenv("RESUME_FAIL", "1")
    proc = run_flow("resume_flow.py", tmp_datastore, "run", check=False)
    assert proc.returncode != 0
    orig_run = latest_run_id(tmp_datastore, "ResumeFlow")

    monkeypatch.setenv("RESUME_FAIL", "0")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))) + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    flows_dir = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                             "flows")
    cmd = [sys.executable, os.path.join(flows_dir, "resume_flow.py"),
           "--datastore-root", tmp_datastore, "resume",
           "--origin-run-id", orig_run, "--reentrant"]
    procs = [subprocess.Popen(cmd, env=env, stdout=subprocess.PIPE,
                              stderr=subprocess.PIPE, text=True)
             for _ in range(2)]
    rcs = [p.wait(timeout=300) for p in procs]
    assert rcs == [0, 0], [(p.stdout.read(), p.stderr.read())
                           for p in procs]
    # middle ran exactly twice total (once in failed run + once in resume)
    assert int(open(counter_dir / "middle").read()) == 2
    assert int(open(counter_dir / "end").read()) == 1


def test_exit_disallow_retry(tmp_datastore, tmp_path):
    """A task exiting with EXIT_DISALLOW_RETRY (202) must NOT be retried
    even under @retry (reference METAFLOW_EXIT_DISALLOW_RETRY contract)."""
    import subprocess
    import sys

    flow = tmp_path / "disallow_flow.py"
    flow.write_text(
        "import os, sys\n"
        "from metaflow_amd import FlowSpec, step, retry, current\n"
        "class DisallowFlow(FlowSpec):\n"
        "    @retry(times=3)\n"
        "    @step\n"
        "    def start(self):\n"
        "        with open(os.environ['ATTEMPT_LOG'], 'a') as f:\n"
        "            f.write('attempt %d\\n' % current.retry_count)\n"
        "        os._exit(202)\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def end(self):\n"
        "        pass\n"
        "if __name__ == '__main__':\n"
        "    DisallowFlow()\n")
    log = tmp_path / "attempts.log"
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    env["ATTEMPT_LOG"] = str(log)
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        capture_output=True, text=True, env=env, timeout=180)
    assert proc.returncode != 0
    # exactly ONE attempt despite @retry(times=3)
    assert log.read_text().count("attempt") == 1, log.read_text()


def test_gang_member_failure_and_resume(tmp_datastore):
    """A dead gang member fails the whole gang; resume reruns the full
    gang (all-or-nothing clone semantics) and the run completes."""
    proc = run_flow("gang_fail_flow.py", tmp_datastore, "run",
                    check=False, env_extra={"GANG_FAIL": "1"})
    assert proc.returncode != 0
    proc2 = run_flow("gang_fail_flow.py", tmp_datastore, "resume",
                     env_extra={"GANG_FAIL": "0"})
    assert proc2.returncode == 0
    run_id = latest_run_id(tmp_datastore, "GangFailFlow")
    assert read_artifact(tmp_datastore, "GangFailFlow", run_id, "join",
                         "ranks") == [0, 1]


def test_empty_foreach_fails_cleanly(tmp_datastore, tmp_path):
    """foreach over an empty sequence raises a clear InvalidNext error
    at the split step (reference behavior: runtime rejects it)."""
    import subprocess
    import sys

    flow = tmp_path / "empty_flow.py"
    flow.write_text(
        "from metaflow_amd import FlowSpec, step\n"
        "class EmptyForeachFlow(FlowSpec):\n"
        "    @step\n"
        "    def start(self):\n"
        "        self.items = []\n"
        "        self.next(self.work, foreach='items')\n"
        "    @step\n"
        "    def work(self):\n"
        "        self.next(self.join)\n"
        "    @step\n"
        "    def join(self, inputs):\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def end(self):\n"
        "        pass\n"
        "if __name__ == '__main__':\n"
        "    EmptyForeachFlow()\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, str(flow), "--datastore-root",
         tmp_datastore, "run"],
        capture_output=True, text=True, env=env, timeout=180)
    assert proc.returncode != 0
    assert "foreach over an empty sequence" in (proc.stdout + proc.stderr)


def test_bad_switch_value_fails_cleanly(tmp_datastore, tmp_path):
    """A switch condition naming a non-existent target step fails with a
    clear error, not a hang or KeyError."""
    import subprocess
    import sys

    flow = tmp_path / "badswitch_flow.py"
    flow.write_text(
        "from metaflow_amd import FlowSpec, step\n"
        "class BadSwitchFlow(FlowSpec):\n"
        "    @step\n"
        "    def start(self):\n"
        "        self.route = 'nowhere'\n"
        "        self.next(self.a, self.b, condition='route')\n"
        "    @step\n"
        "    def a(self):\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def b(self):\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def end(self):\n"
        "        pass\n"
        "if __name__ == '__main__':\n"
        "    BadSwitchFlow()\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, str(flow), "--datastore-root",
         tmp_datastore, "run"],
        capture_output=True, text=True, env=env, timeout=180)
    assert proc.returncode != 0
    out = proc.stdout + proc.stderr
    assert "nowhere" in out or "condition" in out


def test_resume_partial_foreach(tmp_datastore):
    """Resume after ONE foreach child failed: succeeded children clone
    (metadata 'cloned_from'), only the failed child re-executes
    (reference resume_foreach_inner behavior)."""
    import json

    proc = run_flow("foreach_fail_flow.py", tmp_datastore, "run",
                    check=False, env_extra={"FF_FAIL": "1"})
    assert proc.returncode != 0
    proc2 = run_flow("foreach_fail_flow.py", tmp_datastore, "resume",
                     env_extra={"FF_FAIL": "0"})
    assert proc2.returncode == 0
    run_id = latest_run_id(tmp_datastore, "ForeachFailFlow")
    assert read_artifact(tmp_datastore, "ForeachFailFlow", run_id, "join",
                         "total") == 60
    # 3 of the 4 work tasks were cloned, 1 re-executed
    meta_dir = os.path.join(tmp_datastore, "ForeachFailFlow", "_meta",
                            run_id)
    cloned = 0
    for fn in os.listdir(meta_dir):
        if fn.startswith("task.work."):
            info = json.load(open(os.path.join(meta_dir, fn)))
            blob = json.dumps(info)
            if "cloned_from" in blob:
                cloned += 1
    assert cloned == 3, cloned


def test_resume_preserves_parameters(tmp_datastore, tmp_path):
    """resume reuses the ORIGIN run's parameter values (reference
    semantics) — not CLI defaults."""
    import subprocess
    import sys

    flow = tmp_path / "param_resume_flow.py"
    flow.write_text(
        "import os\n"
        "from metaflow_amd import FlowSpec, Parameter, step\n"
        "class ParamResumeFlow(FlowSpec):\n"
        "    scale = Parameter('scale', default=1, type=int)\n"
        "    @step\n"
        "    def start(self):\n"
        "        self.v = self.scale * 7\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def end(self):\n"
        "        if os.environ.get('PR_FAIL') == '1':\n"
        "            raise RuntimeError('planned')\n"
        "        self.final = self.v + self.scale\n"
        "if __name__ == '__main__':\n"
        "    ParamResumeFlow()\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    env["PR_FAIL"] = "1"
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run", "--scale", "9"],
        capture_output=True, text=True, env=env, timeout=180)
    assert proc.returncode != 0
    env["PR_FAIL"] = "0"
    proc2 = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "resume"],
        capture_output=True, text=True, env=env, timeout=180)
    assert proc2.returncode == 0, proc2.stderr[-1500:]
    run_id = latest_run_id(tmp_datastore, "ParamResumeFlow")
    assert read_artifact(tmp_datastore, "ParamResumeFlow", run_id, "end",
                         "final") == 9 * 7 + 9


def test_resume_partial_nested_foreach(tmp_datastore):
    """One failed leaf inside a 2x3 nested foreach: resume clones the 5
    succeeded leaves (and the completed inner join of the other branch),
    reruns only the failed leaf and its dependent joins."""
    import json

    proc = run_flow("nested_foreach_fail_flow.py", tmp_datastore, "run",
                    check=False, env_extra={"NF_FAIL": "1"})
    assert proc.returncode != 0
    proc2 = run_flow("nested_foreach_fail_flow.py", tmp_datastore,
                     "resume", env_extra={"NF_FAIL": "0"})
    assert proc2.returncode == 0
    run_id = latest_run_id(tmp_datastore, "NestedForeachFailFlow")
    assert read_artifact(tmp_datastore, "NestedForeachFailFlow", run_id,
                         "join_o", "total") == 306
    meta_dir = os.path.join(tmp_datastore, "NestedForeachFailFlow",
                            "_meta", run_id)
    cloned = 0
    for fn in os.listdir(meta_dir):
        if fn.startswith("task.leaf."):
            blob = json.dumps(json.load(
                open(os.path.join(meta_dir, fn))))
            if "cloned_from" in blob:
                cloned += 1
    assert cloned == 5, cloned  # the failing leaf sleeps so siblings finish


def test_catch_in_foreach(tmp_datastore):
    """A caught foreach-child failure still reaches the join as a
    FailureHandledByCatch artifact; the run completes."""
    run_flow("catch_foreach_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "CatchForeachFlow")
    assert read_artifact(tmp_datastore, "CatchForeachFlow", run_id,
                         "join", "total") == 20
    fails = read_artifact(tmp_datastore, "CatchForeachFlow", run_id,
                          "join", "failures")
    assert len(fails) == 1 and "boom" in fails[0]


def test_timeout_decorator(tmp_datastore, tmp_path):
    """@timeout(seconds=2) kills a hung step quickly; with @catch the
    flow still completes and records the timeout."""
    import subprocess
    import sys
    import time

    flow = tmp_path / "timeout_flow.py"
    flow.write_text(
        "import time\n"
        "from metaflow_amd import FlowSpec, catch, step, timeout\n"
        "class TimeoutFlow(FlowSpec):\n"
        "    @catch(var='err')\n"
        "    @timeout(seconds=2)\n"
        "    @step\n"
        "    def start(self):\n"
        "        time.sleep(60)\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def end(self):\n"
        "        assert self.err\n"
        "        self.caught = str(self.err)\n"
        "if __name__ == '__main__':\n"
        "    TimeoutFlow()\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    t0 = time.time()
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        capture_output=True, text=True, env=env, timeout=180)
    assert proc.returncode == 0, proc.stderr[-1500:]
    assert time.time() - t0 < 40, "timeout did not fire promptly"
    run_id = latest_run_id(tmp_datastore, "TimeoutFlow")
    caught = read_artifact(tmp_datastore, "TimeoutFlow", run_id, "end",
                           "caught")
    assert "imeout" in caught or "imed out" in caught, caught


def test_resume_step_to_rerun_transitive(tmp_datastore, tmp_path,
                                         monkeypatch):
    """resume --step-to-rerun middle must also rerun downstream steps:
    cloning `end` from the origin would resurrect artifacts computed
    from the PRE-rerun middle output (advisor finding r1 #1; reference
    runtime.py:415-419 expands steps_to_rerun transitively)."""
    counter_dir = tmp_path / "counters"
    counter_dir.mkdir()
    monkeypatch.setenv("RESUME_COUNTER_DIR", str(counter_dir))
    monkeypatch.setenv("RESUME_FAIL", "0")
    run_flow("resume_flow.py", tmp_datastore, "run")
    orig_run = latest_run_id(tmp_datastore, "ResumeFlow")

    run_flow("resume_flow.py", tmp_datastore, "resume",
             "--origin-run-id", orig_run, "--step-to-rerun", "middle")
    new_run = latest_run_id(tmp_datastore, "ResumeFlow")
    assert new_run != orig_run
    # start cloned (ran once); middle AND end re-ran (twice each)
    assert int(open(counter_dir / "start").read()) == 1
    assert int(open(counter_dir / "middle").read()) == 2
    assert int(open(counter_dir / "end").read()) == 2
    assert read_artifact(tmp_datastore, "ResumeFlow", new_run, "end",
                         "z") == 11


def test_catch_on_foreach_split_rejected(tmp_datastore, tmp_path):
    """@catch on a foreach SPLIT step is rejected at graph init: a
    swallowed failure there cannot synthesize a foreach transition
    (no split size), so downstream join bookkeeping would corrupt
    (advisor finding r1 #3; reference catch_decorator.py:45-52)."""
    flow = tmp_path / "catch_split_flow.py"
    flow.write_text(
        "from metaflow_amd import FlowSpec, catch, step\n\n"
        "class CatchSplitFlow(FlowSpec):\n"
        "    @catch(var='err')\n"
        "    @step\n"
        "    def start(self):\n"
        "        self.items = [1, 2]\n"
        "        self.next(self.work, foreach='items')\n\n"
        "    @step\n"
        "    def work(self):\n"
        "        self.next(self.join)\n\n"
        "    @step\n"
        "    def join(self, inputs):\n"
        "        self.next(self.end)\n\n"
        "    @step\n"
        "    def end(self):\n"
        "        pass\n\n"
        "if __name__ == '__main__':\n"
        "    CatchSplitFlow()\n")
    import subprocess
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.run(
        [sys.executable, str(flow), "--datastore-root", tmp_datastore,
         "run"], env=env, capture_output=True, text=True, timeout=120)
    assert proc.returncode != 0
    assert "not supported" in proc.stderr + proc.stdout


def test_gang_transient_failure_retries(tmp_datastore, tmp_path,
                                        monkeypatch):
    """A rank dying on the first attempt (rendezvous-class crash) tears
    the gang down and the WHOLE gang retries once on a fresh port; the
    run completes without resume."""
    marker_dir = tmp_path / "markers"
    marker_dir.mkdir()
    proc = run_flow("gang_retry_flow.py", tmp_datastore, "run",
                    env_extra={"GANG_RETRY_DIR": str(marker_dir)})
    assert proc.returncode == 0, proc.stderr[-3000:]
    run_id = latest_run_id(tmp_datastore, "GangRetryFlow")
    assert read_artifact(tmp_datastore, "GangRetryFlow", run_id, "join",
                         "ranks") == [0, 1]
    # every rank attempted twice (whole-gang retry, not per-rank)
    assert (marker_dir / "attempted_0").read_text() == "xx"
    assert (marker_dir / "attempted_1").read_text() == "xx"


def test_wedged_gang_killed_and_retried(tmp_datastore, tmp_path):
    """A gang rank that hangs forever (simulated RCCL rendezvous wedge)
    is killed by the scheduler's stall detector, the gang tears down
    and retries as a unit, and the retry completes (VERDICT r1 weak #6:
    'a hung gang rank blocks execute() forever')."""
    import time as _time

    marker_dir = tmp_path / "markers"
    marker_dir.mkdir()
    t0 = _time.time()
    proc = run_flow("wedged_gang_flow.py", tmp_datastore, "run",
                    env_extra={"WEDGE_DIR": str(marker_dir),
                               "MFX_TASK_STALL_TIMEOUT": "10"})
    elapsed = _time.time() - t0
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert elapsed < 240, "stall kill took too long (%.0fs)" % elapsed
    assert "MFX_TASK_STALL_TIMEOUT" in proc.stdout + proc.stderr
    run_id = latest_run_id(tmp_datastore, "WedgedGangFlow")
    assert read_artifact(tmp_datastore, "WedgedGangFlow", run_id, "join",
                         "ranks") == [0, 1]
