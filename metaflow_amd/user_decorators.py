"""User step decorators: pure-Python step wrappers with pre/post/skip
semantics.

Parity target: /root/reference/metaflow/user_decorators/user_step_decorator.py
(UserStepDecorator :448 — pre_step/post_step/skip_step contract — and the
@user_step_decorator generator helper :585); execution composes with the
plugin-decorator wrapper chain in task.py (reference task.py:67
_exec_step_function). FlowMutator/MutableFlow (class-rewriting mutators)
live in user_config.py.

Usage — subclass form::

    class Timing(UserStepDecorator):
        def init(self, **kwargs):
            self.label = kwargs.get("label", "t")
        def pre_step(self, step_name, flow, inputs=None):
            self._t0 = time.time()          # or return a replacement fn
        def post_step(self, step_name, flow, exception=None):
            flow.elapsed = time.time() - self._t0
            return exception                # None swallows it

    @Timing(label="train")
    @step
    def start(self): ...

Usage — generator form::

    @user_step_decorator
    def timing(step_name, flow, inputs, attributes):
        t0 = time.time()
        yield                               # run the wrapped step
        flow.elapsed = time.time() - t0

yield nothing -> the step is skipped (its default self.next() is
synthesized from the static graph); yield a callable -> it replaces the
step; yield a dict -> skip with those self.next() kwargs; catch the
exception around the yield -> the step is treated as successful.
"""

import inspect

from .exceptions import MFXException

#: sentinel: skip the wrapped step, synthesizing its default transition
USER_SKIP_STEP = {}


class UserStepDecorator(object):
    """Base class for user step wrappers; subclasses auto-register by
    name (class name, lowercased, unless ``name`` is set)."""

    name = None
    _registry = {}

    def __init_subclass__(cls, **kwargs):
        super().__init_subclass__(**kwargs)
        cls.name = cls.name or cls.__name__.lower()
        UserStepDecorator._registry[cls.name] = cls

    def __new__(cls, *args, **kwargs):
        if len(args) == 1 and not kwargs and callable(args[0]) \
                and getattr(args[0], "is_step", False):
            # bare form: @MyDeco directly above @step. Attach and hand
            # the STEP FUNCTION back (returning a non-instance skips
            # __init__, so the class attribute stays the function).
            inst = object.__new__(cls)
            inst.attributes = {}
            inst._skip_step = False
            inst.init()
            inst._attach(args[0])
            return args[0]
        return object.__new__(cls)

    def __init__(self, *args, **kwargs):
        # parameterized form: @MyDeco(arg=...) above @step
        self.attributes = dict(kwargs)
        self._skip_step = False
        self.init(*args, **kwargs)

    def __call__(self, *args, **kwargs):
        if len(args) != 1 or not getattr(args[0], "is_step", False):
            raise MFXException(
                "@%s must be applied above @step" % type(self).__name__)
        return self._attach(args[0])

    def _attach(self, func):
        if not hasattr(func, "user_wrappers"):
            func.user_wrappers = []
        # python applies decorators bottom-up, so append order is
        # innermost-first — exactly the nesting order apply_user_wrappers
        # wants
        func.user_wrappers.append(self)
        return func

    # -------------------------------------------------------------- hooks
    def init(self, *args, **kwargs):
        """Per-application constructor (the reference replaces __init__
        with this so the base can own the attach dance)."""

    def pre_step(self, step_name, flow, inputs=None):
        """Runs before anything wrapped by this decorator. Return None to
        proceed normally, or a callable with the step's signature to run
        INSTEAD of the wrapped code. May set self.skip_step."""
        return None

    def post_step(self, step_name, flow, exception=None):
        """Runs after the wrapped code (before artifacts persist, so it
        may mutate flow). Return the exception to (re)raise — None
        swallows a failure and the step is considered successful — or a
        tuple (exception, next_kwargs) where next_kwargs synthesizes the
        skipped/failed step's self.next() ({} = graph defaults)."""
        return exception

    @property
    def skip_step(self):
        """False, True (skip with graph-default next), or a dict of
        self.next() kwargs to use when skipping."""
        return self._skip_step

    @skip_step.setter
    def skip_step(self, value):
        self._skip_step = value

    # ---------------------------------------------------------- execution
    def _execute(self, inner, step_name, flow, inputs, graph):
        skip = self.skip_step
        if skip is False or skip is None:
            replacement = self.pre_step(step_name, flow, inputs)
            skip = self.skip_step  # pre_step may have set it
        else:
            replacement = None
        if skip is not False and skip is not None:
            kw = skip if isinstance(skip, dict) else {}
            _synthesize_next(flow, graph, step_name, kw)
            self.post_step(step_name, flow, None)
            return
        target = replacement if replacement is not None else inner
        exc = None
        try:
            _call_steplike(target, flow, inputs)
        except Exception as e:  # noqa: BLE001 — handed to post_step
            exc = e
        res = self.post_step(step_name, flow, exc)
        if isinstance(res, tuple):
            new_exc, next_kwargs = res
        else:
            new_exc, next_kwargs = res, None
        if new_exc is not None:
            raise new_exc
        if flow._transition is None and (exc is not None
                                         or next_kwargs is not None):
            _synthesize_next(flow, graph, step_name, next_kwargs or {})


def _call_steplike(fn, flow, inputs):
    if inputs is None:
        return fn(flow)
    return fn(flow, inputs)


def _synthesize_next(flow, graph, step_name, user_kwargs):
    """Build the self.next() call the step would have made, from the
    static graph, merged with user-provided kwargs."""
    node = graph[step_name]
    if node.type == "end" or not node.out_funcs:
        return
    kwargs = {}
    if node.type == "foreach" and node.foreach_param:
        kwargs["foreach"] = node.foreach_param
    elif node.type == "split-switch" and node.condition:
        kwargs["condition"] = node.condition
    elif node.type == "split-parallel" and node.num_parallel:
        kwargs["num_parallel"] = node.num_parallel
    kwargs.update(user_kwargs or {})
    targets = [getattr(flow, f) for f in node.out_funcs]
    flow.next(*targets, **kwargs)


def apply_user_wrappers(func, wrappers, step_name, graph):
    """Nest the wrapper stack around the (already plugin-decorated) step
    callable; wrappers is innermost-first (attach order)."""
    inner = func
    for w in wrappers:
        def make(w, inner):
            def wrapped(flow, inputs=None):
                w._execute(inner, step_name, flow, inputs, graph)
            return wrapped
        inner = make(w, inner)
    return inner


# ======================= generator-function helper =========================
class _GeneratorStepDecorator(UserStepDecorator):
    """Adapter: a generator function drives pre (before yield), the
    wrapped code (at yield) and post (after yield)."""

    name = "_generator_base"
    _gen_func = None

    def _execute(self, inner, step_name, flow, inputs, graph):
        gf = type(self)._gen_func
        n_params = len(inspect.signature(gf).parameters)
        args = (step_name, flow, inputs)
        if n_params >= 4:
            args = args + (self.attributes,)
        gen = gf(*args)
        try:
            directive = next(gen)
        except StopIteration:
            # never yielded: skip the step entirely
            _synthesize_next(flow, graph, step_name, {})
            return
        if isinstance(directive, dict):
            _synthesize_next(flow, graph, step_name, directive)
            _finish_gen(gen)
            return
        target = directive if callable(directive) else inner
        try:
            result = _call_steplike(target, flow, inputs)
        except Exception as e:  # noqa: BLE001 — offered to the generator
            try:
                gen.throw(e)
            except StopIteration:
                pass  # swallowed: step is successful
            except Exception as e2:  # re-raised (or new) -> step fails
                raise e2
            else:
                _finish_gen(gen)
            if flow._transition is None:
                _synthesize_next(flow, graph, step_name, {})
            return
        # replacement-callable return protocol (reference :625-636)
        if callable(directive) and flow._transition is None:
            if result is True:
                _synthesize_next(flow, graph, step_name, {})
            elif isinstance(result, dict):
                _synthesize_next(flow, graph, step_name, result)
        _finish_gen(gen)


def _finish_gen(gen):
    try:
        next(gen)
    except StopIteration:
        return
    raise MFXException("a @user_step_decorator generator must yield at "
                       "most once")


def user_step_decorator(func=None):
    """Turn a generator function into a user step decorator (see module
    docstring for the yield protocol)."""

    def build(gf):
        if not inspect.isgeneratorfunction(gf):
            raise MFXException("@user_step_decorator requires a "
                               "generator function (it must yield)")
        return type(gf.__name__, (_GeneratorStepDecorator,),
                    {"_gen_func": staticmethod(gf), "name": gf.__name__})

    if func is not None:
        return build(func)
    return build
