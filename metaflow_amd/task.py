"""In-subprocess task execution.

Parity target: /root/reference/metaflow/task.py (MetaflowTask.run_step :570,
_init_foreach :317, _exec_step_function :67). A task process:

1. registers the attempt with metadata,
2. builds input datastores from --input-paths,
3. reconstructs the foreach stack,
4. seeds the output datastore index from its input (artifact passdown),
5. runs decorator hooks + the user step function,
6. persists artifacts + the transition, writes attempt_ok, commits DONE.

The scheduler only ever learns about the task through its exit code and the
datastore/metadata state it leaves behind.
"""

import os
import sys
import traceback

from .current import current, Parallel
from .exceptions import MFXException
from .flowspec import ForeachFrame
from .unbounded_foreach import UBF_CONTROL, UBF_TASK

PARAMETERS_STEP = "_parameters"
PARAMETERS_TASK_ID = "0"


class Inputs(object):
    """The `inputs` object handed to join steps: iterable, indexable, and
    attribute-addressable by step name."""

    def __init__(self, task_inputs):
        self._inputs = task_inputs

    def __iter__(self):
        return iter(self._inputs)

    def __len__(self):
        return len(self._inputs)

    def __getitem__(self, idx):
        return self._inputs[idx]

    def __getattr__(self, name):
        matches = [i for i in self._inputs if i._step_name == name]
        if len(matches) == 1:
            return matches[0]
        raise AttributeError(
            "Join inputs have %d tasks from step '%s'"
            % (len(matches), name))


class TaskInput(object):
    """Read-only view of one input task: artifacts as attributes."""

    def __init__(self, task_ds):
        object.__setattr__(self, "_ds", task_ds)
        object.__setattr__(self, "_step_name", task_ds.step_name)

    def __getattr__(self, name):
        ds = object.__getattribute__(self, "_ds")
        if name in ds:
            return ds[name]
        raise AttributeError(
            "Input task %s has no artifact '%s'" % (ds.pathspec, name))

    def __contains__(self, name):
        return name in object.__getattribute__(self, "_ds")

    def _artifact_names(self):
        return object.__getattribute__(self, "_ds").artifact_names()

    def _artifact_sha(self, name):
        return object.__getattribute__(self, "_ds").artifact_sha(name)

    def _get_artifact(self, name):
        return object.__getattribute__(self, "_ds")[name]

    @property
    def pathspec(self):
        return object.__getattribute__(self, "_ds").pathspec

    @property
    def foreach_stack_frames(self):
        ds = object.__getattribute__(self, "_ds")
        meta = ds.load_metadata("foreach_stack") or []
        return [ForeachFrame(*f) for f in meta]

    def __repr__(self):
        return "TaskInput(%s)" % self.pathspec


class MFXTask(object):
    def __init__(self, flow, graph, flow_datastore, metadata, environment=None):
        self.flow = flow
        self.graph = graph
        self.flow_datastore = flow_datastore
        self.metadata = metadata

    # ------------------------------------------------------------------ utils
    def _parse_pathspec(self, pathspec):
        parts = pathspec.split("/")
        # run/step/task
        if len(parts) == 3:
            return parts
        # flow/run/step/task
        if len(parts) == 4:
            return parts[1:]
        raise MFXException("Bad input pathspec %r" % pathspec)

    def _input_datastores(self, input_paths):
        """Joins with many inputs init their datastores concurrently
        (reference task.py:252: parallel prefetch if >4 inputs)."""
        specs = [self._parse_pathspec(p) for p in input_paths]
        if len(specs) <= 4:
            return [self.flow_datastore.get_task_datastore(r, s, t)
                    for r, s, t in specs]
        from concurrent.futures import ThreadPoolExecutor

        with ThreadPoolExecutor(max_workers=8) as pool:
            return list(pool.map(
                lambda spec: self.flow_datastore.get_task_datastore(*spec),
                specs))

    def _reconstruct_foreach_stack(self, node, input_dss, split_index):
        """Build this task's foreach frame stack from its first input's
        persisted stack (reference task.py:317-415)."""
        if not input_dss:
            return []
        parent = input_dss[0]
        parent_frames = parent.load_metadata("foreach_stack") or []
        stack = [ForeachFrame(*f) for f in parent_frames]

        if node.type == "join":
            split_name = getattr(node, "matching_join_of", None)
            if split_name is None and node.split_parents:
                split_name = node.split_parents[-1]
            split_node = self.graph[split_name] if split_name else None
            if split_node is not None and split_node.type in (
                    "foreach", "split-parallel"):
                stack = stack[:-1]  # pop the foreach frame
            return stack

        # was the parent a foreach/parallel split? then we are one of its
        # children and need a new frame
        ptrans = parent.load_metadata("transition") or {}
        if ptrans.get("foreach") is not None or \
                ptrans.get("num_parallel") is not None:
            if split_index is None:
                raise MFXException(
                    "Step %s is a foreach child but no --split-index was "
                    "given." % node.name)
            stack.append(ForeachFrame(
                parent.step_name,
                ptrans.get("foreach"),
                ptrans.get("num_splits"),
                int(split_index),
            ))
        return stack

    # ------------------------------------------------------------------- run
    def run_step(self, step_name, run_id, task_id, input_paths, split_index,
                 retry_count, max_user_code_retries, origin_run_id=None,
                 ubf_context=None, namespace=None):
        from .profile_util import from_start

        flow = self.flow
        node = self.graph[step_name]
        # unbound class function: decorators wrap it, then we call f(flow)
        step_func = getattr(type(flow), step_name)
        decorators = getattr(step_func, "decorators", [])

        from_start("task: enter run_step")
        self.metadata.register_task(run_id, step_name, task_id, retry_count,
                                    metadata={"attempt_started": True})
        from_start("task: attempt registered")

        # gang rank CPU pinning (set by the gang scheduler; the NUMA
        # half of HIP_VISIBLE_DEVICES)
        aff = os.environ.get("MFX_CPU_AFFINITY")
        if aff and hasattr(os, "sched_setaffinity"):
            try:
                lo, hi = aff.split("-")
                os.sched_setaffinity(0, range(int(lo), int(hi) + 1))
            except (ValueError, OSError):
                pass

        # task-level liveness (reference task.py:797: heartbeats for
        # task+run); lossy sidecar, never blocks the task
        hb_sidecar = None
        try:
            from .sidecar import SidecarSubProcess

            hb_sidecar = SidecarSubProcess("heartbeat", {
                "flow_name": flow.name,
                "run_id": run_id,
                "datastore_root": self.flow_datastore.datastore_root,
                "provider": getattr(self.metadata, "TYPE", "local"),
                "step_name": step_name,
                "task_id": str(task_id),
            })
        except Exception:
            pass

        from_start("task: sidecar up")
        input_dss = self._input_datastores(input_paths)
        from_start("task: input datastores built")
        output = self.flow_datastore.get_task_datastore(
            run_id, step_name, task_id, attempt=retry_count, mode="w")
        output.init_task()

        # ---- foreach stack --------------------------------------------------
        stack = self._reconstruct_foreach_stack(node, input_dss, split_index)
        output.save_metadata("foreach_stack", [list(f) for f in stack])

        # ---- artifact passdown ---------------------------------------------
        if node.type == "join":
            # joins only pass down parameters (+ internals); everything else
            # must go through merge_artifacts
            param_names = set()
            if input_dss:
                pn = input_dss[0].get("_parameter_names", [])
                param_names = set(pn) | {"_parameter_names"}
            for ds in input_dss[:1]:
                output.passdown(ds, names=param_names)
        elif input_dss:
            output.passdown(input_dss[0])

        # ---- flow instance state -------------------------------------------
        flow._datastore = output
        flow._foreach_stack = stack
        flow._transition = None
        flow._current_step = step_name
        from .flowspec import _NOT_SET

        object.__setattr__(flow, "_cached_input", _NOT_SET)
        if "_parameter_names" in output:
            from .includefile import IncludedFile

            flow._parameter_names = output["_parameter_names"]
            lazy_includes = {}
            for pname in flow._parameter_names:
                if pname in output:
                    value = output[pname]
                    if isinstance(value, IncludedFile):
                        # content stays in the CAS until first access
                        # (FlowSpec.__getattr__ decodes on demand)
                        lazy_includes[pname] = value
                    else:
                        object.__setattr__(flow, pname, value)
            object.__setattr__(flow, "_lazy_includes", lazy_includes)

        # ---- gang context ---------------------------------------------------
        parallel_ctx = None
        if os.environ.get("MFX_PARALLEL_NUM_NODES"):
            parallel_ctx = Parallel(
                main_ip=os.environ.get("MFX_PARALLEL_MAIN_IP", "127.0.0.1"),
                main_port=int(os.environ.get("MFX_PARALLEL_MAIN_PORT", "0")),
                num_nodes=int(os.environ["MFX_PARALLEL_NUM_NODES"]),
                node_index=int(os.environ.get("MFX_PARALLEL_NODE_INDEX",
                                              "0")),
                control_task_id=os.environ.get(
                    "MFX_PARALLEL_CONTROL_TASK_ID", str(task_id)),
            )

        current._set_env(
            flow_name=flow.name,
            run_id=run_id,
            step_name=step_name,
            task_id=str(task_id),
            retry_count=retry_count,
            origin_run_id=origin_run_id,
            namespace=namespace,
            username=os.environ.get("USER"),
            is_running=True,
        )
        if parallel_ctx is not None:
            current._update_env({"parallel": parallel_ctx})
        from .plugins.trigger_decorator import TriggerInfo

        trig = TriggerInfo.from_env()
        if trig is not None:
            current._update_env({"trigger": trig})

        # ---- join inputs ----------------------------------------------------
        inputs = None
        if node.type == "join":
            inputs = Inputs([TaskInput(ds) for ds in input_dss])

        # ---- unbounded-foreach control task ---------------------------------
        # (reference protocol: plugins/test_unbounded_foreach_decorator.py:99
        # + runtime.py:1178-1264 — the control task spawns/waits mappers and
        # persists _control_mapper_tasks for the join)
        if ubf_context == UBF_CONTROL:
            return self._run_ubf_control(
                flow, node, output, input_dss, run_id, step_name, task_id,
                retry_count)

        # ---- decorator hooks + user code ------------------------------------
        from .monitor import get_system_logger, get_system_monitor

        monitor = get_system_monitor()
        get_system_logger().log({
            "event": "task_start", "flow": flow.name, "run_id": run_id,
            "step": step_name, "task_id": task_id, "attempt": retry_count})
        task_ok = True
        error = None
        try:
            for deco in decorators:
                deco.task_pre_step(
                    step_name, output, self.metadata, run_id, task_id, flow,
                    self.graph, retry_count, max_user_code_retries,
                    ubf_context, inputs)

            func = step_func
            for deco in decorators:
                func = deco.task_decorate(
                    func, flow, self.graph, retry_count,
                    max_user_code_retries, ubf_context)
            # user step wrappers (pre/post/skip semantics) nest OUTSIDE
            # the plugin-decorator chain (reference task.py:67)
            user_wrappers = getattr(step_func, "user_wrappers", None)
            if user_wrappers:
                from .user_decorators import apply_user_wrappers

                func = apply_user_wrappers(func, user_wrappers,
                                           step_name, self.graph)

            with monitor.measure("mfx.task.user_code"), \
                    monitor.count("mfx.task.runs"):
                if node.type == "join":
                    self._exec_step_function(func, flow, inputs)
                else:
                    self._exec_step_function(func, flow)

            for deco in decorators:
                deco.task_post_step(step_name, flow, self.graph, retry_count,
                                    max_user_code_retries)
        except Exception as ex:
            task_ok = False
            error = ex
            tb = traceback.format_exc()
            sys.stderr.write(tb)
            swallowed = False
            for deco in decorators:
                try:
                    if deco.task_exception(ex, step_name, flow, self.graph,
                                           retry_count,
                                           max_user_code_retries):
                        swallowed = True
                except Exception:
                    pass
            if swallowed:
                task_ok = True
                error = None

        # ---- validate + persist transition ----------------------------------
        transition = flow._transition
        if task_ok and node.type != "end":
            if transition is None:
                task_ok = False
                error = MFXException(
                    "Step %s did not call self.next()." % step_name)
            else:
                bad = [f for f in transition["out_funcs"]
                       if f not in node.out_funcs]
                if bad:
                    task_ok = False
                    error = MFXException(
                        "Step %s transitioned to %s which does not match "
                        "the static graph (expected %s)."
                        % (step_name, bad, node.out_funcs))

        # ---- persist ---------------------------------------------------------
        from_start("task: step function done")
        try:
            flow._task_ok = task_ok
            if error is not None:
                flow._exception = "%s: %s" % (type(error).__name__, error)
            output.persist(flow)
        except Exception:
            task_ok = False
            traceback.print_exc()
        from_start("task: artifacts persisted")

        if transition is not None:
            output.save_metadata("transition", transition)
        output.save_metadata("attempt_ok", {
            "ok": task_ok,
            "error": (str(error) if error else None),
        })
        self.metadata.register_metadata(
            run_id, step_name, task_id, retry_count,
            {"attempt_ok": task_ok})

        for deco in decorators:
            try:
                deco.task_finished(step_name, flow, self.graph, task_ok,
                                   retry_count, max_user_code_retries)
            except Exception:
                traceback.print_exc()

        if hb_sidecar is not None:
            hb_sidecar.terminate()
        from_start("task: sidecar terminated")
        if task_ok:
            output.done()
            from_start("task: DONE committed")
        else:
            raise TaskFailed(error)

    def _run_ubf_control(self, flow, node, output, input_dss, run_id,
                         step_name, task_id, retry_count):
        """Control side of an unbounded foreach: enumerate the UBF input,
        spawn one mapper subprocess per item (same step, task ids
        <ctrl>_mapper_<i>), wait, persist _control_mapper_tasks."""
        import subprocess

        from .config import MAX_WORKERS

        parent = input_dss[0]
        ptrans = parent.load_metadata("transition") or {}
        var = ptrans.get("foreach")
        seq = getattr(flow, var)
        try:
            num = len(seq)
        except TypeError:
            num = len(list(iter(seq)))

        def mapper_cmd(i, mapper_id):
            cmd = list(sys.argv)
            cmd.insert(0, sys.executable)

            def replace(flag, value):
                if flag in cmd:
                    cmd[cmd.index(flag) + 1] = value
                else:
                    cmd.extend([flag, value])

            replace("--task-id", mapper_id)
            replace("--split-index", str(i))
            replace("--ubf-context", UBF_TASK)
            return cmd

        mapper_ids = ["%s_mapper_%d" % (task_id, i) for i in range(num)]
        procs = {}
        failed = []
        i = 0
        while i < num or procs:
            while i < num and len(procs) < MAX_WORKERS:
                procs[i] = subprocess.Popen(mapper_cmd(i, mapper_ids[i]))
                i += 1
            done = []
            for idx, proc in procs.items():
                rc = proc.poll()
                if rc is not None:
                    done.append(idx)
                    if rc != 0:
                        failed.append(idx)
            for idx in done:
                del procs[idx]
            if not done:
                import time

                time.sleep(0.05)
        if failed:
            raise TaskFailed(MFXException(
                "UBF mappers %s failed." % failed))

        mapper_paths = ["%s/%s/%s" % (run_id, step_name, mid)
                        for mid in mapper_ids]
        flow._control_mapper_tasks = mapper_paths
        flow._transition = {
            "out_funcs": list(node.out_funcs),
            "foreach": None,
            "condition": None,
            "num_parallel": None,
            "num_splits": None,
        }
        flow._task_ok = True
        output.persist(flow)
        output.save_metadata("transition", flow._transition)
        output.save_metadata("control_mapper_tasks", mapper_paths)
        output.save_metadata("attempt_ok", {"ok": True, "control": True})
        self.metadata.register_metadata(run_id, step_name, task_id,
                                        retry_count, {"attempt_ok": True})
        output.done()

    @staticmethod
    def _exec_step_function(func, flow, inputs=None):
        # func may be the unbound class function or already bound/wrapped
        if inputs is None:
            func(flow)
        else:
            func(flow, inputs)


class TaskFailed(MFXException):
    headline = "Task failed"

    def __init__(self, error):
        super().__init__(str(error))
        self.error = error


def load_task_metadata_transition(task_ds):
    """Scheduler helper: read a finished task's transition without
    unpickling artifacts."""
    return task_ds.load_metadata("transition")


def dump_parameters(flow_datastore, run_id, values, graph_info=None):
    """Persist the `_parameters` pseudo-task for a run."""
    ds = flow_datastore.get_task_datastore(
        run_id, PARAMETERS_STEP, PARAMETERS_TASK_ID, attempt=0, mode="w")
    ds.init_task()
    arts = [("_parameter_names", sorted(values))]
    for k, v in values.items():
        arts.append((k, v))
    if graph_info is not None:
        arts.append(("_graph_info", graph_info))
    ds.save_artifacts(arts)
    ds.save_metadata("foreach_stack", [])
    ds.save_metadata("attempt_ok", {"ok": True})
    ds.done()
    return "%s/%s/%s" % (run_id, PARAMETERS_STEP, PARAMETERS_TASK_ID)
