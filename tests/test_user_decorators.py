"""UserStepDecorator subsystem: wrap, replace, skip, ordering, exception
swallowing, generator form — run through real subprocess-driven flows
(reference user_decorators/user_step_decorator.py semantics)."""

import os
import subprocess
import sys
import textwrap

from .test_runtime import REPO, latest_run_id, read_artifact


def _run_inline_flow(tmp_path, tmp_datastore, body, name="udeco_flow.py",
                     check=True, env_extra=None):
    flow = tmp_path / name
    flow.write_text(textwrap.dedent(body))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    env.update(env_extra or {})
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet",
         "--datastore-root", tmp_datastore, "run"],
        env=env, capture_output=True, text=True, timeout=300)
    if check:
        assert proc.returncode == 0, proc.stderr[-3000:]
    return proc


def test_wrap_pre_post_and_ordering(tmp_path, tmp_datastore):
    """pre/post hooks fire around the step; stacked wrappers nest with
    the decorator closest to @step innermost."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        from metaflow_amd import FlowSpec, UserStepDecorator, step

        class Tag(UserStepDecorator):
            def init(self, **kw):
                self.tag = kw["tag"]
            def pre_step(self, step_name, flow, inputs=None):
                flow.trace = getattr(flow, "trace", []) + \\
                    ["pre-" + self.tag]
            def post_step(self, step_name, flow, exception=None):
                flow.trace = flow.trace + ["post-" + self.tag]
                return exception

        class OuterTag(Tag):
            pass

        class FlowA(FlowSpec):
            @OuterTag(tag="outer")
            @Tag(tag="inner")
            @step
            def start(self):
                self.trace = self.trace + ["body"]
                self.next(self.end)

            @step
            def end(self):
                assert self.trace == [
                    "pre-outer", "pre-inner", "body",
                    "post-inner", "post-outer"], self.trace
                self.ok = True

        if __name__ == "__main__":
            FlowA()
    """)
    run_id = latest_run_id(tmp_datastore, "FlowA")
    assert read_artifact(tmp_datastore, "FlowA", run_id, "end", "ok")


def test_replace_step(tmp_path, tmp_datastore):
    """pre_step returning a callable replaces the step body."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        from metaflow_amd import FlowSpec, UserStepDecorator, step

        class Replace(UserStepDecorator):
            def pre_step(self, step_name, flow, inputs=None):
                def other(flow_obj):
                    flow_obj.who = "replacement"
                    flow_obj.next(flow_obj.end)
                return other

        class FlowB(FlowSpec):
            @Replace
            @step
            def start(self):
                self.who = "original"
                self.next(self.end)

            @step
            def end(self):
                assert self.who == "replacement", self.who

        if __name__ == "__main__":
            FlowB()
    """)


def test_skip_step(tmp_path, tmp_datastore):
    """skip_step=True skips the body and synthesizes the default
    self.next() from the static graph."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        from metaflow_amd import FlowSpec, UserStepDecorator, step

        class Skip(UserStepDecorator):
            def pre_step(self, step_name, flow, inputs=None):
                self.skip_step = True

        class FlowC(FlowSpec):
            @step
            def start(self):
                self.ran = ["start"]
                self.next(self.middle)

            @Skip
            @step
            def middle(self):
                self.ran = self.ran + ["middle"]
                self.next(self.end)

            @step
            def end(self):
                assert self.ran == ["start"], self.ran
                self.ok = True

        if __name__ == "__main__":
            FlowC()
    """)
    run_id = latest_run_id(tmp_datastore, "FlowC")
    assert read_artifact(tmp_datastore, "FlowC", run_id, "end", "ok")


def test_swallow_exception(tmp_path, tmp_datastore):
    """post_step returning None swallows the step's exception; the
    wrapper records it as an artifact and the run completes."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        from metaflow_amd import FlowSpec, UserStepDecorator, step

        class Guard(UserStepDecorator):
            def post_step(self, step_name, flow, exception=None):
                flow.guarded = repr(exception) if exception else None
                return None  # swallow

        class FlowD(FlowSpec):
            @Guard
            @step
            def start(self):
                raise ValueError("boom")
                self.next(self.end)

            @step
            def end(self):
                assert "boom" in self.guarded, self.guarded
                self.ok = True

        if __name__ == "__main__":
            FlowD()
    """)
    run_id = latest_run_id(tmp_datastore, "FlowD")
    assert read_artifact(tmp_datastore, "FlowD", run_id, "end", "ok")


def test_generator_form(tmp_path, tmp_datastore):
    """@user_step_decorator generator: code before/after the yield runs
    around the step; attributes arrive as the 4th argument."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        from metaflow_amd import FlowSpec, step, user_step_decorator

        @user_step_decorator
        def stamp(step_name, flow, inputs, attributes):
            flow.pre_mark = attributes.get("mark", "?")
            yield
            flow.post_mark = flow.pre_mark + "-done"

        class FlowE(FlowSpec):
            @stamp(mark="m1")
            @step
            def start(self):
                assert self.pre_mark == "m1"
                self.next(self.end)

            @step
            def end(self):
                assert self.post_mark == "m1-done", self.post_mark
                self.ok = True

        if __name__ == "__main__":
            FlowE()
    """)
    run_id = latest_run_id(tmp_datastore, "FlowE")
    assert read_artifact(tmp_datastore, "FlowE", run_id, "end", "ok")


def test_generator_skip_and_catch(tmp_path, tmp_datastore):
    """Generator protocol: yielding a dict skips the step; catching the
    exception around the yield makes a failing step successful."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        from metaflow_amd import FlowSpec, step, user_step_decorator

        @user_step_decorator
        def skipper(step_name, flow, inputs):
            yield {}

        @user_step_decorator
        def catcher(step_name, flow, inputs):
            try:
                yield
            except RuntimeError as e:
                flow.caught = str(e)

        class FlowF(FlowSpec):
            @step
            def start(self):
                self.ran = []
                self.next(self.skipped)

            @skipper
            @step
            def skipped(self):
                self.ran = self.ran + ["skipped"]
                self.next(self.fails)

            @catcher
            @step
            def fails(self):
                raise RuntimeError("caught-me")
                self.next(self.end)

            @step
            def end(self):
                assert self.ran == [], self.ran
                assert self.caught == "caught-me", self.caught
                self.ok = True

        if __name__ == "__main__":
            FlowF()
    """)
    run_id = latest_run_id(tmp_datastore, "FlowF")
    assert read_artifact(tmp_datastore, "FlowF", run_id, "end", "ok")


def test_wrapper_composes_with_catch_and_foreach(tmp_path, tmp_datastore):
    """A user wrapper on a foreach CHILD composes with @catch: the
    wrapper runs per-iteration and @catch still swallows a failure."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        from metaflow_amd import FlowSpec, UserStepDecorator, catch, step

        class Count(UserStepDecorator):
            def pre_step(self, step_name, flow, inputs=None):
                flow.counted = True

        class FlowG(FlowSpec):
            @step
            def start(self):
                self.items = [0, 1, 2]
                self.next(self.work, foreach="items")

            @catch(var="err")
            @Count
            @step
            def work(self):
                assert self.counted
                if self.input == 1:
                    raise ValueError("boom")
                self.val = self.input * 2
                self.next(self.join)

            @step
            def join(self, inputs):
                self.total = sum(getattr(i, "val", 0) for i in inputs)
                self.n_failed = sum(
                    1 for i in inputs if getattr(i, "err", None))
                self.next(self.end)

            @step
            def end(self):
                assert self.total == 4, self.total
                assert self.n_failed == 1

        if __name__ == "__main__":
            FlowG()
    """)


def test_user_wrapper_with_resume(tmp_path, tmp_datastore):
    """User wrappers re-run on resumed (non-cloned) steps and stay
    silent on cloned ones; composes with IncludeFile lazy decode."""
    data = tmp_path / "inc.txt"
    data.write_text("alpha\nbeta\n")
    marker = tmp_path / "wrapped_runs"
    flow = tmp_path / "wr_flow.py"
    flow.write_text('''
import os

from metaflow_amd import FlowSpec, IncludeFile, UserStepDecorator, step


class Count(UserStepDecorator):
    def pre_step(self, step_name, flow, inputs=None):
        with open(os.environ["WRAP_MARKER"], "a") as f:
            f.write(step_name + "\\n")


class WRFlow(FlowSpec):
    notes = IncludeFile("notes", required=True)

    @step
    def start(self):
        self.n = len(self.notes.splitlines())
        self.next(self.mid)

    @Count
    @step
    def mid(self):
        if os.environ.get("WR_FAIL") == "1":
            raise RuntimeError("fail first")
        self.m = self.n * 2
        self.next(self.end)

    @step
    def end(self):
        assert self.m == 4, self.m


if __name__ == "__main__":
    WRFlow()
''')
    import subprocess
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    env["WRAP_MARKER"] = str(marker)
    env["WR_FAIL"] = "1"
    p1 = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run", "--notes", str(data)],
        env=env, capture_output=True, text=True, timeout=300)
    assert p1.returncode != 0
    env["WR_FAIL"] = "0"
    p2 = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "resume"],
        env=env, capture_output=True, text=True, timeout=300)
    assert p2.returncode == 0, p2.stderr[-3000:]
    # wrapper ran on the failed attempt AND the resumed re-run of mid
    assert marker.read_text().count("mid") == 2


def test_step_mutator(tmp_path, tmp_datastore):
    """StepMutator (reference user_flow_decorator.py analog): placed
    above @step, its mutate(MutableStep) runs at config-resolution
    time and can attach real decorators (here @retry) to that step."""
    _run_inline_flow(tmp_path, tmp_datastore, """
        import os

        from metaflow_amd import FlowSpec, StepMutator, step

        class AddRetry(StepMutator):
            def mutate(self, ms):
                from metaflow_amd.plugins.retry_decorator import (
                    RetryDecorator,
                )
                ms.add_decorator(RetryDecorator, times=2)

        class SMFlow(FlowSpec):
            @AddRetry()
            @step
            def start(self):
                self.marker = os.path.join(
                    os.environ["SM_DIR"], "tries")
                with open(self.marker, "a") as f:
                    f.write("x")
                if len(open(self.marker).read()) < 2:
                    raise ValueError("flaky")
                self.tries = len(open(self.marker).read())
                self.next(self.end)

            @step
            def end(self):
                assert self.tries == 2, self.tries
                self.ok = True

        if __name__ == "__main__":
            SMFlow()
    """, env_extra={"SM_DIR": str(tmp_path)})
    run_id = latest_run_id(tmp_datastore, "SMFlow")
    assert read_artifact(tmp_datastore, "SMFlow", run_id, "end", "ok")


def test_default_namespace_and_metadata_helpers(tmp_datastore,
                                                monkeypatch):
    """default_namespace / get_metadata / default_metadata parity
    helpers."""
    monkeypatch.setenv("MFX_DATASTORE_ROOT", tmp_datastore)
    import importlib

    import metaflow_amd.client as C
    importlib.reload(C)
    from metaflow_amd.client import (
        default_metadata,
        default_namespace,
        get_metadata,
        get_namespace,
        namespace,
    )

    namespace(None)
    assert get_namespace() is None
    ns = default_namespace()
    assert ns.startswith("user:") and get_namespace() == ns
    md = get_metadata()
    assert md.startswith("local@")
    assert default_metadata() == md
