"""Per-task ambient context, importable as ``from metaflow_amd import current``.

Parity target: /root/reference/metaflow/metaflow_current.py (Current,
Parallel). Decorators may inject additional properties via
``current._update_env``.
"""

from collections import namedtuple

Parallel = namedtuple(
    "Parallel",
    ["main_ip", "main_port", "num_nodes", "node_index", "control_task_id"],
)


class Current(object):
    def __init__(self):
        self._env = {}
        self._flow_name = None
        self._run_id = None
        self._step_name = None
        self._task_id = None
        self._retry_count = 0
        self._origin_run_id = None
        self._namespace = None
        self._username = None
        self._is_running = False
        self._tags = set()

    def _set_env(
        self,
        flow_name=None,
        run_id=None,
        step_name=None,
        task_id=None,
        retry_count=0,
        origin_run_id=None,
        namespace=None,
        username=None,
        is_running=True,
        tags=None,
    ):
        self._flow_name = flow_name
        self._run_id = run_id
        self._step_name = step_name
        self._task_id = task_id
        self._retry_count = retry_count
        self._origin_run_id = origin_run_id
        self._namespace = namespace
        self._username = username
        self._is_running = is_running
        self._tags = set(tags or ())

    def _update_env(self, env_vars):
        for k, v in env_vars.items():
            self._env[k] = v

    def __getattr__(self, name):
        # only called for attrs not found normally; check injected props
        env = object.__getattribute__(self, "_env")
        if name in env:
            return env[name]
        raise AttributeError(name)

    def __contains__(self, key):
        return getattr(self, key, None) is not None

    def get(self, key, default=None):
        return getattr(self, key, default)

    @property
    def is_running_flow(self):
        return self._is_running

    @property
    def flow_name(self):
        return self._flow_name

    @property
    def run_id(self):
        return self._run_id

    @property
    def step_name(self):
        return self._step_name

    @property
    def task_id(self):
        return self._task_id

    @property
    def retry_count(self):
        return self._retry_count

    @property
    def origin_run_id(self):
        return self._origin_run_id

    @property
    def namespace(self):
        return self._namespace

    @property
    def username(self):
        return self._username

    @property
    def tags(self):
        return self._tags

    @property
    def pathspec(self):
        if None in (self._flow_name, self._run_id, self._step_name,
                    self._task_id):
            return None
        return "/".join(
            (self._flow_name, self._run_id, self._step_name, self._task_id)
        )


current = Current()
