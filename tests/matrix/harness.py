"""Matrix test harness: cartesian product of DAG shapes x test behaviors,
each generating a real flow file, running it through the CLI, and
validating with two independent checkers (CLI dump + client API).

Parity target: the reference's test/core harness (SURVEY §4 tier 1:
run_tests.py + FlowFormatter + CliCheck/MetadataCheck).
"""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

# ---------------------------------------------------------------- graphs
# node: (name, kind, targets, extra)
# kinds: linear | split | foreach | join | parallel | end
GRAPHS = {
    "linear": [
        ("start", "linear", ["a"], {}),
        ("a", "linear", ["b"], {}),
        ("b", "linear", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "branch": [
        ("start", "split", ["a", "b"], {}),
        ("a", "linear", ["join_ab"], {}),
        ("b", "linear", ["join_ab"], {}),
        ("join_ab", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "nested_branch": [
        ("start", "split", ["a", "b"], {}),
        ("a", "split", ["aa", "ab"], {}),
        ("aa", "linear", ["join_a"], {}),
        ("ab", "linear", ["join_a"], {}),
        ("join_a", "join", ["join_all"], {}),
        ("b", "linear", ["bb"], {}),
        ("bb", "linear", ["join_all"], {}),
        ("join_all", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "foreach": [
        ("start", "foreach", ["inner"], {"var": "fanout", "n": 3}),
        ("inner", "linear", ["join_f"], {}),
        ("join_f", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "nested_foreach": [
        ("start", "foreach", ["mid"], {"var": "fanout", "n": 2}),
        ("mid", "foreach", ["inner"], {"var": "fanout2", "n": 2}),
        ("inner", "linear", ["join_i"], {}),
        ("join_i", "join", ["join_o"], {}),
        ("join_o", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "branch_in_foreach": [
        ("start", "foreach", ["mid"], {"var": "fanout", "n": 2}),
        ("mid", "split", ["x", "y"], {}),
        ("x", "linear", ["join_xy"], {}),
        ("y", "linear", ["join_xy"], {}),
        ("join_xy", "join", ["join_f"], {}),
        ("join_f", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "parallel": [
        ("start", "parallel", ["work"], {"n": 2}),
        ("work", "linear", ["join_p"], {}),
        ("join_p", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "switch": [
        ("start", "switch", ["fast", "slow"], {"var": "route"}),
        ("fast", "linear", ["merge"], {}),
        ("slow", "linear", ["merge"], {}),
        ("merge", "linear", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "ubf": [
        ("start", "ubf", ["inner"], {"var": "payload", "n": 3}),
        ("inner", "linear", ["join_u"], {}),
        ("join_u", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    # composite switch placements (reference switch_in_foreach /
    # branch_in_switch graphs): only the 'fast' arm executes
    "switch_in_foreach": [
        ("start", "foreach", ["mid"], {"var": "fanout", "n": 2}),
        ("mid", "switch", ["fast", "slow"], {"var": "route"}),
        ("fast", "linear", ["conv"], {}),
        ("slow", "linear", ["conv"], {}),
        ("conv", "linear", ["join_f"], {}),
        ("join_f", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "branch_in_switch": [
        ("start", "switch", ["fast", "slow"], {"var": "route"}),
        ("fast", "split", ["fa", "fb"], {}),
        ("fa", "linear", ["join_fab"], {}),
        ("fb", "linear", ["join_fab"], {}),
        ("join_fab", "join", ["conv"], {}),
        ("slow", "linear", ["conv"], {}),
        ("conv", "linear", ["end"], {}),
        ("end", "end", [], {}),
    ],
    "foreach_in_switch": [
        ("start", "switch", ["fast", "slow"], {"var": "route"}),
        ("fast", "foreach", ["item"], {"var": "fanout", "n": 3}),
        ("item", "linear", ["join_i"], {}),
        ("join_i", "join", ["conv"], {}),
        ("slow", "linear", ["conv"], {}),
        ("conv", "linear", ["end"], {}),
        ("end", "end", [], {}),
    ],
    # scheduler stress: a long sequential chain (transition latency adds
    # up; artifact passdown must stay metadata-only the whole way)
    "deep_linear": [
        ("start", "linear", ["s1"], {}),
    ] + [
        ("s%d" % i, "linear", ["s%d" % (i + 1)], {}) for i in range(1, 8)
    ] + [
        ("s8", "linear", ["end"], {}),
        ("end", "end", [], {}),
    ],
    # scheduler stress: one split fanning into 6 concurrent branches
    # under the worker cap, all gated by a single join
    "wide_branch": [
        ("start", "split", ["w%d" % i for i in range(6)], {}),
    ] + [
        ("w%d" % i, "linear", ["join_w"], {}) for i in range(6)
    ] + [
        ("join_w", "join", ["end"], {}),
        ("end", "end", [], {}),
    ],
}


class MatrixTest(object):
    """Subclass and define code snippets per qualifier. Qualifiers:
    'start', 'end', 'join', 'parallel-step', 'foreach-inner', 'all'
    (every non-join step). check(run) validates via the client API."""

    def body(self, name, kind, graph_name):
        """Return code lines for the given step, before its transition."""
        return ["pass"]

    def check(self, run, graph):
        pass

    def execute(self, flow_file, env, datastore_root):
        """Run the generated flow; behaviors may override (e.g. resume)."""
        return cli(flow_file, env, datastore_root, "run")


def cli(flow_file, env, datastore_root, *args, extra_env=None):
    e = dict(env)
    if extra_env:
        e.update(extra_env)
    return subprocess.run(
        [sys.executable, flow_file, "--quiet", "--datastore-root",
         datastore_root] + list(args),
        capture_output=True, text=True, env=e, timeout=300)


def generate_flow(graph_name, test, class_name):
    graph = GRAPHS[graph_name]
    lines = [
        "import os",
        "from metaflow_amd import FlowSpec, step, current, parallel",
        "from metaflow_amd import UnboundedForeachInput",
        "",
        "class MXListUBF(UnboundedForeachInput):",
        "    def __init__(self, items):",
        "        self.items = list(items)",
        "    def __iter__(self):",
        "        return iter(self.items)",
        "    def __len__(self):",
        "        return len(self.items)",
        "    def __getitem__(self, i):",
        "        return self if i is None else self.items[i]",
        "",
        "class %s(FlowSpec):" % class_name,
    ]
    for name, kind, targets, extra in graph:
        body = test.body(name, kind, graph_name) or ["pass"]
        args = "self" if kind != "join" else "self, inputs"
        lines.append("    @step")
        lines.append("    def %s(%s):" % (name, args))
        if kind == "foreach":
            lines.append("        self.%s = list(range(%d))"
                         % (extra["var"], extra["n"]))
        elif kind == "ubf":
            lines.append("        self.%s = MXListUBF(range(%d))"
                         % (extra["var"], extra["n"]))
        elif kind == "switch":
            lines.append("        self.%s = '%s'"
                         % (extra["var"], "fast"))
        for b in body:
            lines.append("        " + b)
        # transition
        if kind == "end":
            pass
        elif kind in ("foreach", "ubf"):
            lines.append("        self.next(self.%s, foreach='%s')"
                         % (targets[0], extra["var"]))
        elif kind == "switch":
            lines.append("        self.next(%s, condition='%s')"
                         % (", ".join("self.%s" % t for t in targets),
                            extra["var"]))
        elif kind == "parallel":
            lines.append("        self.next(self.%s, num_parallel=%d)"
                         % (targets[0], extra["n"]))
        elif kind == "split":
            lines.append("        self.next(%s)"
                         % ", ".join("self.%s" % t for t in targets))
        else:
            lines.append("        self.next(self.%s)" % targets[0])
        lines.append("")
    lines.append("if __name__ == '__main__':")
    lines.append("    %s()" % class_name)
    return "\n".join(lines)


def run_matrix_case(graph_name, test, tmp_dir, datastore_root,
                    executor="cli"):
    """executor: "cli" re-invokes the flow file's CLI (the scheduler's
    own process model); "api" drives the same file through the Runner
    (reference contexts run both executors per case)."""
    class_name = "MX%s%sFlow" % (
        graph_name.title().replace("_", ""), type(test).__name__)
    src = generate_flow(graph_name, test, class_name)
    flow_file = os.path.join(tmp_dir, "%s.py" % class_name)
    with open(flow_file, "w") as f:
        f.write(src)
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    if executor == "api":
        os.environ["MFX_NUM_GPUS"] = "0"
        os.environ["PYTHONPATH"] = env["PYTHONPATH"]
        from metaflow_amd.runner import Runner

        ex = Runner(flow_file, datastore_root=datastore_root).run()
        assert ex.status == "successful", ex.status
    else:
        proc = test.execute(flow_file, env, datastore_root)
        if proc.returncode != 0:
            raise AssertionError(
                "matrix flow %s x %s failed:\n%s\n%s"
                % (graph_name, type(test).__name__, proc.stdout[-3000:],
                   proc.stderr[-3000:]))

    # checker 1: client API
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = datastore_root
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    run = client.Flow(class_name).latest_run
    assert run.successful
    test.check(run, GRAPHS[graph_name])

    # checker 2: CLI dump of the end task must succeed and show artifacts
    end_task = run["end"].task
    dump = subprocess.run(
        [sys.executable, flow_file, "--quiet", "--datastore-root",
         datastore_root, "dump",
         "/".join(end_task.pathspec.split("/")[1:])],
        capture_output=True, text=True, env=env, timeout=120)
    assert dump.returncode == 0, dump.stderr[-2000:]
    return run, dump.stdout


# ------------------------------------------------------------ behaviors
class ArtifactFlow(MatrixTest):
    """Artifacts set at start propagate to every non-join step and merge
    through joins untouched (set-once semantics)."""

    def body(self, name, kind, graph_name):
        if name == "start":
            return ["self.base = 42", "self.text = 'hello' * 10"]
        if kind == "join":
            return ["self.merge_artifacts(inputs)"]
        if kind == "end":
            return ["assert self.base == 42", "assert len(self.text) == 50"]
        return ["assert self.base == 42"]

    def check(self, run, graph):
        assert run["end"].task.data.base == 42
        assert run["end"].task.data.text == "hello" * 10


class StepCounterFlow(MatrixTest):
    """Every step contributes a distinct artifact; joins merge them all;
    end checks the full set arrived."""

    def body(self, name, kind, graph_name):
        if kind == "join":
            return ["self.merge_artifacts(inputs)"]
        lines = ["self.mark_%s = '%s'" % (name, name)]
        if kind == "end":
            lines.append("assert self.mark_start == 'start'")
        return lines

    def check(self, run, graph):
        data = run["end"].task.data
        # every non-join step's mark must have propagated to the end;
        # switch graphs only run the chosen branch ('fast' here)
        skipped = {t for n, k, ts, _e in graph if k == "switch"
                   for t in ts if t != "fast"}
        for name, kind, _t, _e in graph:
            if kind != "join" and name not in skipped:
                assert getattr(data, "mark_%s" % name) == name


class CurrentInfoFlow(MatrixTest):
    """current.* is coherent in every task; gang steps see
    current.parallel (reference basic_parallel)."""

    def body(self, name, kind, graph_name):
        if kind == "join":
            return [
                "assert current.step_name == '%s'" % name,
                "self.gang_idx = sorted(i.node_idx for i in inputs "
                "if hasattr(i, 'node_idx'))",
                "self.merge_artifacts(inputs, "
                "exclude=['seen_step', 'node_idx'])",
            ]
        lines = [
            "assert current.flow_name == type(self).__name__",
            "assert current.step_name == '%s'" % name,
            "assert current.run_id is not None",
            "self.seen_step = current.step_name",
        ]
        if graph_name == "parallel" and name == "work":
            lines += [
                "assert current.parallel.num_nodes == 2",
                "assert 0 <= current.parallel.node_index < 2",
                "assert current.parallel.main_ip",
                "self.node_idx = current.parallel.node_index",
            ]
        return lines

    def check(self, run, graph):
        assert run["end"].task.data.seen_step == "end"
        names = [n for n, _k, _t, _e in graph]
        if "join_p" in names:
            assert run["join_p"].task.data.gang_idx == [0, 1]


class ForeachContextFlow(MatrixTest):
    """self.index/self.input are correct inside fan-outs."""

    def body(self, name, kind, graph_name):
        if name == "inner" and "foreach" in graph_name:
            return [
                "assert self.index is not None",
                "assert self.input == self.index",
                "self.seen_index = self.index",
            ]
        if kind == "join":
            return [
                "self.all_idx = sorted(getattr(i, 'seen_index', -1) "
                "for i in inputs)",
            ]
        return ["pass"]

    def check(self, run, graph):
        names = [n for n, k, _t, _e in graph]
        if "join_f" in names and "inner" in names:
            t = run["join_f"].task
            if "mid" not in names:  # plain foreach x3
                assert t.data.all_idx == [0, 1, 2], t.data.all_idx


class ResumeFlow(MatrixTest):
    """Clone-based resume across every DAG shape: the first run fails at
    `end` (env MX_FAIL=1); `resume` must clone the whole successful
    prefix (foreach fan-outs, joins, gangs included) and rerun only the
    failed step."""

    def body(self, name, kind, graph_name):
        if kind == "join":
            return ["self.merge_artifacts(inputs)"]
        lines = ["self.mark_%s = '%s'" % (name, name)]
        if kind == "end":
            lines += [
                "if os.environ.get('MX_FAIL') == '1':",
                "    raise Exception('planned failure')",
                "self.finished = 1",
            ]
        return lines

    def execute(self, flow_file, env, datastore_root):
        first = cli(flow_file, env, datastore_root, "run",
                    extra_env={"MX_FAIL": "1"})
        assert first.returncode != 0, "first run should fail at end"
        return cli(flow_file, env, datastore_root, "resume",
                   extra_env={"MX_FAIL": "0"})

    def check(self, run, graph):
        data = run["end"].task.data
        assert data.finished == 1
        skipped = {t for n, k, ts, _e in graph if k == "switch"
                   for t in ts if t != "fast"}
        for name, kind, _t, _e in graph:
            if kind != "join" and name not in skipped:
                assert getattr(data, "mark_%s" % name) == name
        # the resumed run must contain tasks for every cloned step too
        for name, _k, _t, _e in graph:
            if name in skipped:
                continue
            assert len(list(run[name])) >= 1, "no tasks for %s" % name


class NumpyArtifactFlow(MatrixTest):
    """A numpy artifact created at start is LOADED in every non-join
    step (exercising the numpy-v1 zero-copy codec + provenance-skip
    persist across every DAG shape): the deserialized view is
    read-only, and the SAME blob sha propagates to the end
    unchanged (no re-serialization drift through splits, foreach,
    gangs or joins)."""

    def body(self, name, kind, graph_name):
        if name == "start":
            return ["import numpy as np",
                    "self.arr = np.arange(4096, dtype=np.int32)"]
        if kind == "join":
            return ["self.merge_artifacts(inputs)"]
        return ["assert int(self.arr[:4].sum()) == 6",
                "assert not self.arr.flags.writeable"]

    def check(self, run, graph):
        import numpy as np

        end_task = run["end"].task
        a = end_task.data.arr
        assert isinstance(a, np.ndarray) and a.shape == (4096,)
        assert int(a.sum()) == 4095 * 4096 // 2
        start_sha = run["start"].task["arr"].sha
        assert end_task["arr"].sha == start_sha, \
            "numpy artifact re-serialized along the way"
