"""Flow CLI: ``python myflow.py [top-opts] {run,resume,step,show,dump,logs}``.

Parity target: /root/reference/metaflow/cli.py:333 (start group) and
cli_components/{run_cmds,step_cmd,dump_cmd}.py. The CLI is the
inter-process API: the scheduler launches every task as
``python flow.py step <name> --run-id ...`` (reference runtime.py:2344).
"""

import json
import sys
import traceback

import click

from .config import (
    DEFAULT_DATASTORE,
    DEFAULT_METADATA,
    MAX_NUM_SPLITS,
    MAX_WORKERS,
)
from .datastore import FlowDataStore, STORAGE_IMPLS
from .datastore.storage import LocalStorage
from .exceptions import MFXException
from .graph import FlowGraph
from .lint import lint
from .metadata import METADATA_PROVIDERS
from .task import MFXTask


class CLIState(object):
    def __init__(self, flow):
        self.flow = flow
        self.flow_cls = type(flow)
        self.graph = None
        self.flow_datastore = None
        self.metadata = None
        self.quiet = False
        self.top_level_args = []


def _make_storage(datastore, datastore_root):
    impl = STORAGE_IMPLS.get(datastore)
    if impl is None:
        raise MFXException("Unknown datastore '%s' (known: %s)"
                           % (datastore, ", ".join(STORAGE_IMPLS)))
    root = datastore_root
    if root is None and impl is LocalStorage:
        root = LocalStorage.get_datastore_root_from_config()
    return impl(root)


def _init_state(state, datastore, datastore_root, metadata, quiet, with_,
                config=()):
    from .decorators import attach_decorators
    from .user_config import resolve_configs

    overrides = {}
    for spec in config:
        name, _, path = spec.partition("=")
        overrides[name] = path
    resolve_configs(state.flow_cls, overrides)
    if with_:
        attach_decorators(state.flow_cls, with_)
    state.graph = FlowGraph(state.flow_cls)
    lint(state.graph)
    storage = _make_storage(datastore, datastore_root)
    state.flow_datastore = FlowDataStore(state.flow_cls.__name__, storage)
    provider = METADATA_PROVIDERS.get(metadata)
    if provider is None:
        raise MFXException("Unknown metadata provider '%s'" % metadata)
    state.metadata = provider(state.flow_cls.__name__, storage)
    state.quiet = quiet
    # flow-level decorator init (e.g. @project computes its namespace)
    for deco in getattr(state.flow_cls, "_flow_decorators", []):
        deco.flow_init(state.flow_cls, state.graph, None,
                       state.flow_datastore, state.metadata, None, None,
                       {})
    # args to replay on child processes so they see the same config
    state.top_level_args = [
        "--datastore", datastore,
        "--datastore-root", storage.root,
        "--metadata", metadata,
    ]
    if quiet:
        state.top_level_args.append("--quiet")
    for spec in config:
        state.top_level_args.extend(["--config", spec])
    # replay --with so task subprocesses attach the same decorators
    # (reference: runtime replays top_level_options incl. --with)
    for spec in with_ or ():
        state.top_level_args.extend(["--with", spec])
    # run step_init hooks
    for step_name in state.flow_cls._steps:
        func = getattr(state.flow_cls, step_name)
        for deco in getattr(func, "decorators", []):
            deco.step_init(state.flow_cls, state.graph, step_name,
                           func.decorators, datastore, None)


def _param_options(f, flow_cls):
    for name, param in reversed(flow_cls._params):
        f = click.option(
            "--" + name.replace("_", "-"),
            "param_%s" % name,
            default=None,
            required=False,
            help=param.help or "Parameter %s" % name,
        )(f)
    return f


def _resolve_params(flow_cls, kwargs):
    values = {}
    for name, param in flow_cls._params:
        raw = kwargs.get("param_%s" % name)
        if raw is None:
            v = param.resolve_default()
            if v is None and param.required:
                raise MFXException("Parameter '%s' is required." % name)
            values[name] = param.convert(v) if v is not None else None
        else:
            values[name] = param.convert(raw)
    return values


def main(flow):
    state = CLIState(flow)
    flow_cls = state.flow_cls

    @click.group(context_settings={"help_option_names": ["-h", "--help"]})
    @click.option("--datastore", default=DEFAULT_DATASTORE,
                  help="Datastore backend.")
    @click.option("--datastore-root", default=None,
                  help="Datastore root path.")
    @click.option("--metadata", default=DEFAULT_METADATA,
                  help="Metadata provider.")
    @click.option("--quiet", is_flag=True, default=False)
    @click.option("--with", "with_", multiple=True,
                  help="Attach a decorator to all steps, e.g. retry:times=2")
    @click.option("--config", multiple=True,
                  help="Override a Config file: name=path.json")
    def cli(datastore, datastore_root, metadata, quiet, with_, config):
        _init_state(state, datastore, datastore_root, metadata, quiet,
                    with_, config)

    def _run_common(kwargs, clone_run_id=None, steps_to_rerun=None):
        from .runtime import NativeRuntime

        values = None
        if clone_run_id is None:
            values = _resolve_params(flow_cls, kwargs)
            # IncludeFile params: upload the file bytes to the CAS once
            # (raw blob, dedup across runs) and keep only the small
            # IncludedFile handle as the parameter artifact
            from .includefile import upload_include_files

            upload_include_files(values, state.flow_datastore)
        runtime = NativeRuntime(
            flow_cls,
            state.graph,
            state.flow_datastore,
            state.metadata,
            flow_file=sys.argv[0],
            run_id=kwargs.get("run_id"),
            param_values=values,
            clone_run_id=clone_run_id,
            steps_to_rerun=steps_to_rerun,
            max_workers=kwargs.get("max_workers") or MAX_WORKERS,
            max_num_splits=kwargs.get("max_num_splits") or MAX_NUM_SPLITS,
            quiet=state.quiet,
            top_level_args=state.top_level_args,
            tags=tuple(kwargs.get("tag") or ()) + tuple(
                getattr(flow_cls, "_project_tags", ())),
        )
        rid_file = kwargs.get("run_id_file")
        if rid_file:
            with open(rid_file, "w") as f:
                f.write(str(runtime.run_id))
        attr_file = kwargs.get("runner_attribute_file")
        if attr_file:
            with open(attr_file, "w") as f:
                json.dump({"flow_name": flow_cls.__name__,
                           "run_id": runtime.run_id,
                           "metadata": "local@%s"
                           % state.flow_datastore.datastore_root}, f)
        success = False
        try:
            runtime.execute()
            success = True
        finally:
            for deco in getattr(flow_cls, "_flow_decorators", []):
                if deco.name == "exit_hook":
                    try:
                        deco.run_hooks(sys.argv[0], success,
                                       runtime.run_id)
                    except Exception:
                        traceback.print_exc()
        return runtime

    @cli.command(help="Run the flow locally.")
    @click.option("--run-id", default=None)
    @click.option("--run-id-file", default=None,
                  help="Write the run id to this file (reference "
                       "run_id_file behavior).")
    @click.option("--max-workers", default=MAX_WORKERS, type=int)
    @click.option("--max-num-splits", default=MAX_NUM_SPLITS, type=int)
    @click.option("--tag", multiple=True)
    @click.option("--runner-attribute-file", default=None)
    @(lambda f: _param_options(f, flow_cls))
    def run(**kwargs):
        from .system_context import SCHEDULER, set_phase

        set_phase(SCHEDULER)
        _run_common(kwargs)

    @cli.command(help="Resume a failed run from where it left off.")
    @click.option("--run-id-file", default=None,
                  help="Write the new run id to this file.")
    @click.option("--origin-run-id", default=None,
                  help="Run to clone from (default: this flow's most "
                       "recent run, reference semantics)")
    @click.option("--run-id", default=None)
    @click.option("--step-to-rerun", default=None,
                  help="Force this step (and everything after) to rerun.")
    @click.option("--max-workers", default=MAX_WORKERS, type=int)
    @click.option("--max-num-splits", default=MAX_NUM_SPLITS, type=int)
    @click.option("--tag", multiple=True)
    @click.option("--runner-attribute-file", default=None)
    @click.option("--reentrant", is_flag=True, default=False,
                  help="Safe under concurrent invocation: one caller wins "
                       "leader election and resumes; the rest wait.")
    def resume(origin_run_id, step_to_rerun, reentrant, **kwargs):
        from .system_context import SCHEDULER, set_phase

        set_phase(SCHEDULER)
        if origin_run_id is None:
            runs = state.metadata.list_runs()
            if not runs:
                raise MFXException(
                    "No previous run of %s to resume." % flow_cls.__name__)
            origin_run_id = sorted(r["run_id"] for r in runs)[-1]
        steps = [step_to_rerun] if step_to_rerun else []
        if reentrant:
            # leader election via exclusive-create in the datastore
            # (reference runtime.py:1660-1786 uses metadata task
            # registration as the lock; the primitive is the same)
            import time as _time
            import uuid

            run_id = kwargs.get("run_id") or "resume%s" % origin_run_id
            kwargs["run_id"] = run_id
            lock_path = state.flow_datastore.storage.path_join(
                flow_cls.__name__, run_id, "_resume_leader.lock")
            token = uuid.uuid4().hex.encode()
            if state.flow_datastore.storage.create_exclusive(lock_path,
                                                             token):
                _run_common(kwargs, clone_run_id=origin_run_id,
                            steps_to_rerun=steps)
            else:
                click.echo("[mfx] another resume holds the leader lock; "
                           "waiting for run %s" % run_id)
                deadline = _time.time() + 24 * 3600
                while _time.time() < deadline:
                    info = state.metadata.get_run(run_id) or {}
                    if info.get("status") == "successful":
                        return
                    if info.get("status") == "failed":
                        raise MFXException("Leader's resume failed.")
                    _time.sleep(2)
                raise MFXException("Timed out waiting for the leader.")
        else:
            _run_common(kwargs, clone_run_id=origin_run_id,
                        steps_to_rerun=steps)

    @cli.command(help="[internal] Execute one task of one step.")
    @click.argument("step_name")
    @click.option("--run-id", required=True)
    @click.option("--task-id", required=True)
    @click.option("--input-paths", default="")
    @click.option("--split-index", default=None, type=int)
    @click.option("--retry-count", default=0, type=int)
    @click.option("--max-user-code-retries", default=0, type=int)
    @click.option("--origin-run-id", default=None)
    @click.option("--ubf-context", default=None)
    def step(step_name, run_id, task_id, input_paths, split_index,
             retry_count, max_user_code_retries, origin_run_id, ubf_context):
        from .system_context import TASK, set_phase

        set_phase(TASK)
        task = MFXTask(state.flow, state.graph, state.flow_datastore,
                       state.metadata)
        paths = [p for p in input_paths.split(",") if p]
        try:
            task.run_step(step_name, run_id, task_id, paths, split_index,
                          retry_count, max_user_code_retries,
                          origin_run_id=origin_run_id,
                          ubf_context=ubf_context)
        except Exception:
            traceback.print_exc()
            sys.exit(1)

    @cli.command(help="Re-run ONE step of a finished run against its "
                      "original input artifacts (debugging).")
    @click.argument("step_name")
    @click.option("--run-id", "origin_run_id", required=True,
                  help="The finished run to spin against.")
    @click.option("--split-index", default=None, type=int,
                  help="Foreach index when spinning a foreach-child step.")
    def spin(step_name, origin_run_id, split_index):
        from .task import PARAMETERS_STEP as PSTEP

        node = state.graph[step_name]
        if node.type == "join" and len(node.in_funcs) > 1:
            in_steps = sorted(node.in_funcs)
        else:
            in_steps = sorted(node.in_funcs) or [PSTEP]
        input_paths = []
        for s in in_steps:
            tasks = state.flow_datastore.list_tasks(origin_run_id, s)
            if not tasks:
                raise MFXException(
                    "Origin run %s has no tasks for step %s"
                    % (origin_run_id, s))
            input_paths.append("%s/%s/%s" % (origin_run_id, s,
                                             sorted(tasks)[0]))
        spin_run_id = "spin%s" % origin_run_id
        task = MFXTask(state.flow, state.graph, state.flow_datastore,
                       state.metadata)
        task.run_step(step_name, spin_run_id, "1", input_paths,
                      split_index, 0, 0)
        click.echo("spin task done: %s/%s/1" % (spin_run_id, step_name))

    @cli.command(name="tag", help="Mutate run tags: tag add/remove RUN_ID "
                                  "TAG [TAG...]")
    @click.argument("action", type=click.Choice(["add", "remove"]))
    @click.argument("run_id")
    @click.argument("tags", nargs=-1, required=True)
    def tag_cmd(action, run_id, tags):
        if action == "add":
            state.metadata.add_run_tags(run_id, list(tags))
        else:
            state.metadata.remove_run_tags(run_id, list(tags))
        click.echo("tags updated")

    @cli.command(help="Show the flow structure.")
    def show():
        click.echo("Flow: %s" % flow_cls.__name__)
        for name in state.graph.sorted_nodes():
            node = state.graph[name]
            click.echo("  %s [%s] -> %s"
                       % (name, node.type, ", ".join(node.out_funcs) or "-"))

    @cli.command(help="Output the flow graph in DOT format.")
    def output_dot():
        click.echo(state.graph.output_dot())

    @cli.command(help="Dump artifacts of a task: PATHSPEC = run/step/task")
    @click.argument("pathspec")
    @click.option("--max-value-size", default=1000, type=int)
    @click.option("--include", default="",
                  help="Comma-separated artifact names.")
    @click.option("--file", "file_", default=None,
                  help="Serialize artifacts into this pickle file.")
    def dump(pathspec, max_value_size, include, file_):
        import pickle

        parts = pathspec.split("/")
        run_id, step_name, task_id = parts[0], parts[1], parts[2]
        ds = state.flow_datastore.get_task_datastore(run_id, step_name,
                                                     task_id)
        names = [n for n in include.split(",") if n] or [
            n for n in ds.artifact_names() if not n.startswith("_")]
        out = {}
        for name, obj in ds.load_artifacts(names):
            out[name] = obj
        if file_:
            with open(file_, "wb") as f:
                pickle.dump({pathspec: out}, f)
        else:
            for name in sorted(out):
                click.echo("%s: %s" % (name, repr(out[name])
                                       [:max_value_size]))

    @cli.command(help="Show logs of a task: PATHSPEC = run/step/task")
    @click.argument("pathspec")
    @click.option("--stderr", is_flag=True, default=False)
    @click.option("--timestamps", is_flag=True, default=False)
    def logs(pathspec, stderr, timestamps):
        from .mflog import parse

        parts = pathspec.split("/")
        ds = state.flow_datastore.get_task_datastore(parts[0], parts[1],
                                                     parts[2])
        raw = ds.load_logs("stderr" if stderr else "stdout")
        for line in raw.splitlines():
            p = parse(line)
            if p is None:
                click.echo(line)
            elif timestamps:
                click.echo("%s %s" % (p.ts, p.msg))
            else:
                click.echo(p.msg)

    cli(standalone_mode=True)
