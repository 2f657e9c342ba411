"""Decorator machinery: lifecycle-hook plugin system for steps and flows.

Parity target: /root/reference/metaflow/decorators.py:115-560 (Decorator,
StepDecorator hooks, FlowDecorator, @step, --with attach). The hook set is
kept; the implementation is fresh and smaller.

StepDecorator lifecycle (in execution order, per task):
    step_init(flow, graph, step_name, decos, datastore_type)   [graph build]
    runtime_init(flow, graph, package, run_id)                 [scheduler]
    runtime_step_cli(args, retry_count, max_retries, ubf_ctx)  [scheduler,
        mutate the child's CLI args/env before launch]
    task_pre_step(...)                                         [child]
    task_decorate(step_func, flow, graph, retry_count, ...)    [child,
        wrap or replace the user step function]
    task_post_step(...) / task_exception(...)                  [child]
    task_finished(...)                                         [child]
    step_task_retry_count() -> (user_retries, error_retries)   [scheduler]
"""

import functools

from .exceptions import MFXException


class BadDecoratorAttribute(MFXException):
    headline = "Bad decorator attribute"


class Decorator(object):
    name = "decorator"
    defaults = {}
    # decorators that may appear at most once per step
    allow_multiple = False

    def __init__(self, attributes=None, statically_defined=True):
        self.attributes = dict(self.defaults)
        self.statically_defined = statically_defined
        if attributes:
            for k, v in attributes.items():
                if k not in self.defaults:
                    raise BadDecoratorAttribute(
                        "Decorator @%s has no attribute '%s' (valid: %s)"
                        % (self.name, k, ", ".join(self.defaults) or "none"))
                self.attributes[k] = v

    @classmethod
    def parse_spec(cls, deco_spec):
        """Parse 'name:key=val,key2=val2' into (name, attrs)."""
        parts = deco_spec.split(":", 1)
        name = parts[0]
        attrs = {}
        if len(parts) > 1 and parts[1]:
            for kv in parts[1].split(","):
                k, _, v = kv.partition("=")
                # try numeric coercion
                for conv in (int, float):
                    try:
                        v = conv(v)
                        break
                    except (ValueError, TypeError):
                        pass
                attrs[k] = v
        return name, attrs

    def make_decorator_spec(self):
        if not self.attributes:
            return self.name
        attrs = ",".join("%s=%s" % (k, v) for k, v in
                         sorted(self.attributes.items()) if v is not None)
        return "%s:%s" % (self.name, attrs) if attrs else self.name

    def __repr__(self):
        return "@%s(%s)" % (self.name, self.attributes)


class StepDecorator(Decorator):
    def step_init(self, flow, graph, step_name, decorators, datastore_type,
                  logger):
        pass

    def package_init(self, flow, step_name, environment):
        pass

    def step_task_retry_count(self):
        """(user_code_retries, error_retries) added to the scheduler's
        retry budget for tasks of this step."""
        return 0, 0

    def runtime_init(self, flow, graph, package, run_id):
        pass

    def runtime_task_created(self, task_datastore, task_id, split_index,
                             input_paths, is_cloned, ubf_context):
        pass

    def runtime_step_cli(self, args, retry_count, max_user_code_retries,
                         ubf_context):
        pass

    def task_pre_step(self, step_name, task_datastore, metadata, run_id,
                      task_id, flow, graph, retry_count,
                      max_user_code_retries, ubf_context, inputs):
        pass

    def task_decorate(self, step_func, flow, graph, retry_count,
                      max_user_code_retries, ubf_context):
        return step_func

    def task_post_step(self, step_name, flow, graph, retry_count,
                       max_user_code_retries):
        pass

    def task_exception(self, exception, step_name, flow, graph, retry_count,
                       max_user_code_retries):
        """Return True to swallow the exception (see @catch)."""
        return False

    def task_finished(self, step_name, flow, graph, is_task_ok, retry_count,
                      max_user_code_retries):
        pass


class FlowDecorator(Decorator):
    def flow_init(self, flow, graph, environment, flow_datastore, metadata,
                  logger, echo, options):
        pass

    def get_top_level_options(self):
        return []


def _base_step(f):
    """The @step decorator: marks a method as a workflow step."""
    f.is_step = True
    f.decorators = getattr(f, "decorators", [])
    f.name = f.__name__

    @functools.wraps(f)
    def wrapper(*args, **kwargs):
        return f(*args, **kwargs)

    wrapper.is_step = True
    wrapper.decorators = f.decorators
    wrapper.__wrapped_step__ = f
    return f  # keep original object so AST lineno mapping stays simple


step = _base_step


def _attach_decorator_instance(func, deco):
    """Prepend (decorators closest to the function apply first)."""
    existing = getattr(func, "decorators", [])
    if not deco.allow_multiple and any(d.name == deco.name
                                       for d in existing):
        raise MFXException(
            "Step '%s' already has decorator @%s"
            % (getattr(func, "name", func.__name__), deco.name))
    func.decorators = [deco] + existing


def make_step_decorator(cls):
    """Build a user-facing decorator from a StepDecorator subclass.

    Supports both bare `@retry` and parameterized `@retry(times=2)` forms.
    """

    def deco(*args, **kwargs):
        if args and callable(args[0]) and not kwargs:
            func = args[0]
            _attach_decorator_instance(func, cls())
            return func

        def wrap(func):
            _attach_decorator_instance(func, cls(attributes=kwargs))
            return func

        return wrap

    deco.decorator_class = cls
    deco.__name__ = cls.name
    return deco


def make_flow_decorator(cls):
    def deco(*args, **kwargs):
        if args and callable(args[0]) and not kwargs:
            flow_cls = args[0]
            flow_cls._flow_decorators = getattr(
                flow_cls, "_flow_decorators", []) + [cls()]
            return flow_cls

        def wrap(flow_cls):
            flow_cls._flow_decorators = getattr(
                flow_cls, "_flow_decorators", []) + [cls(attributes=kwargs)]
            return flow_cls

        return wrap

    deco.decorator_class = cls
    deco.__name__ = cls.name
    return deco


def attach_decorators(flow_cls, deco_specs):
    """Attach decorators from CLI --with specs to every step that does not
    already carry them (reference: decorators.py:744)."""
    from .plugins import STEP_DECORATORS

    for spec in deco_specs:
        name, attrs = Decorator.parse_spec(spec)
        if name not in STEP_DECORATORS:
            raise MFXException(
                "Unknown decorator '%s' in --with (known: %s)"
                % (name, ", ".join(sorted(STEP_DECORATORS))))
        cls = STEP_DECORATORS[name]
        for step_name in flow_cls._steps:
            func = getattr(flow_cls, step_name)
            if not any(d.name == name for d in func.decorators):
                func.decorators = func.decorators + [
                    cls(attributes=attrs, statically_defined=False)]


def decorators_for_step(func):
    return getattr(func, "decorators", [])
