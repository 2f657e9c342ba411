"""Config(...) parameters + flow mutators.

Parity target: /root/reference/metaflow/user_configs/ (ConfigInput,
ConfigValue) and user_decorators/ (FlowMutator, MutableFlow/MutableStep).
Configs resolve BEFORE graph finalization (from JSON files or literal
dicts), so mutators can rewrite the DAG/decorators based on them.
"""

import json
import os

from .exceptions import MFXException


class ConfigValue(object):
    """Read-only dict with attribute access, nested."""

    def __init__(self, data):
        object.__setattr__(self, "_data", dict(data))

    def __getattr__(self, name):
        data = object.__getattribute__(self, "_data")
        if name in data:
            v = data[name]
            return ConfigValue(v) if isinstance(v, dict) else v
        raise AttributeError(name)

    def __getitem__(self, key):
        v = object.__getattribute__(self, "_data")[key]
        return ConfigValue(v) if isinstance(v, dict) else v

    def __contains__(self, key):
        return key in object.__getattribute__(self, "_data")

    def get(self, key, default=None):
        data = object.__getattribute__(self, "_data")
        v = data.get(key, default)
        return ConfigValue(v) if isinstance(v, dict) else v

    def __setattr__(self, name, value):
        raise MFXException("ConfigValue is read-only")

    def to_dict(self):
        return dict(object.__getattribute__(self, "_data"))

    def __repr__(self):
        return "ConfigValue(%r)" % object.__getattribute__(self, "_data")


class Config(object):
    """Class attribute: resolved to a ConfigValue before the flow runs.

        class F(FlowSpec):
            cfg = Config("cfg", default="config.json")
    """

    IS_CONFIG = True

    def __init__(self, name, default=None, default_value=None, parser=None,
                 required=False):
        self.name = name
        self.default = default          # path to a JSON file
        self.default_value = default_value  # literal dict fallback
        self.parser = parser
        self.required = required

    def resolve(self, explicit_path=None):
        path = explicit_path or os.environ.get(
            "MFX_CONFIG_%s" % self.name.upper()) or self.default
        if path and os.path.exists(os.path.expanduser(str(path))):
            with open(os.path.expanduser(str(path))) as f:
                raw = f.read()
            data = self.parser(raw) if self.parser else json.loads(raw)
        elif self.default_value is not None:
            data = dict(self.default_value)
        elif self.required:
            raise MFXException(
                "Config '%s' requires a file (looked for %r)."
                % (self.name, path))
        else:
            data = {}
        return ConfigValue(data)


def resolve_configs(flow_cls, overrides=None):
    """Replace every Config class attribute with its ConfigValue.
    overrides: {config_name: path} from the CLI. The original Config specs
    are kept in a class registry so a later call with overrides (e.g. the
    CLI after an import-time mutator already resolved) re-resolves."""
    overrides = overrides or {}
    specs = dict(getattr(flow_cls, "_config_specs", {}) or {})
    for cls in flow_cls.__mro__:
        for attr_name, attr in list(vars(cls).items()):
            if isinstance(attr, Config) and attr_name not in specs:
                specs[attr_name] = attr
    flow_cls._config_specs = specs
    resolved = {}
    for attr_name, spec in specs.items():
        value = spec.resolve(overrides.get(spec.name))
        setattr(flow_cls, attr_name, value)
        resolved[attr_name] = value
    apply_step_mutators(flow_cls)
    return resolved


# ------------------------------------------------------------- mutators
class MutableStep(object):
    def __init__(self, flow_cls, step_name):
        self.flow_cls = flow_cls
        self.name = step_name
        self._func = getattr(flow_cls, step_name)

    @property
    def decorators(self):
        return list(getattr(self._func, "decorators", []))

    def add_decorator(self, deco_cls, **attrs):
        from .decorators import _attach_decorator_instance

        _attach_decorator_instance(
            self._func, deco_cls(attributes=attrs,
                                 statically_defined=False))

    def remove_decorator(self, name):
        self._func.decorators = [
            d for d in getattr(self._func, "decorators", [])
            if d.name != name]


class MutableFlow(object):
    def __init__(self, flow_cls):
        self.flow_cls = flow_cls

    @property
    def steps(self):
        return [MutableStep(self.flow_cls, s) for s in self.flow_cls._steps]

    def step(self, name):
        return MutableStep(self.flow_cls, name)

    @property
    def configs(self):
        out = {}
        for attr_name in dir(self.flow_cls):
            v = getattr(self.flow_cls, attr_name, None)
            if isinstance(v, ConfigValue):
                out[attr_name] = v
        return out


class StepMutator(object):
    """Per-step analog of FlowMutator (reference user_decorators/
    user_flow_decorator.py StepMutator): subclass, implement
    mutate(mutable_step), place ABOVE @step. The mutation runs when
    configs resolve (cli._init_state), so it can read ConfigValues:

        class Instrument(StepMutator):
            def mutate(self, ms):
                ms.add_decorator(RetryDecorator, times=2)

        class F(FlowSpec):
            @Instrument()
            @step
            def train(self): ...
    """

    def __init__(self, *args, **kwargs):
        self.args = args
        self.kwargs = kwargs

    def mutate(self, mutable_step):
        raise NotImplementedError

    def __call__(self, step_func):
        if not getattr(step_func, "is_step", False):
            raise TypeError("StepMutator must be applied above @step")
        pending = getattr(step_func, "step_mutators", [])
        step_func.step_mutators = pending + [self]
        return step_func


def apply_step_mutators(flow_cls):
    """Run pending StepMutators (resolve_configs calls this after
    config resolution, once per process)."""
    for name in getattr(flow_cls, "_steps", []):
        func = getattr(flow_cls, name, None)
        for m in getattr(func, "step_mutators", []) or []:
            if getattr(m, "_mfx_applied", False):
                continue
            m.mutate(MutableStep(flow_cls, name))
            m._mfx_applied = True


class FlowMutator(object):
    """Subclass and implement mutate(mutable_flow); apply as a class
    decorator:

        class AddRetries(FlowMutator):
            def mutate(self, mf):
                for s in mf.steps:
                    s.add_decorator(RetryDecorator, times=2)

        @AddRetries()
        class MyFlow(FlowSpec): ...
    """

    def __init__(self, *args, **kwargs):
        self.args = args
        self.kwargs = kwargs

    def mutate(self, mutable_flow):
        raise NotImplementedError

    def __call__(self, flow_cls):
        # configs resolve first so mutate() can read them
        resolve_configs(flow_cls)
        self.mutate(MutableFlow(flow_cls))
        return flow_cls
