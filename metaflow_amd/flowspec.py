"""FlowSpec: the user-facing flow base class.

Parity target: /root/reference/metaflow/flowspec.py (FlowSpec :266,
FlowSpecMeta :166, next() :909, merge_artifacts :738, foreach_stack :654,
lazy __getattr__ :599). Fresh implementation: artifacts are instance
attributes, ``self.next()`` records the transition, the static DAG comes from
graph.py's AST pass, and artifact passdown happens at the datastore layer.
"""

from collections import namedtuple

from .exceptions import (
    InvalidNextException,
    MFXException,
    MissingInMergeArtifactsException,
    UnhandledInMergeArtifactsException,
)
from .parameters import Parameter

ForeachFrame = namedtuple("ForeachFrame",
                          ["step", "var", "num_splits", "index"])

# attribute names on the instance that are never treated as artifacts
INTERNAL_ATTRS = frozenset(
    [
        "_datastore",
        "_foreach_stack",
        "_transition",
        "_parameter_names",
        "_graph",
        "_steps_cache",
        "_current_step",
        "_cached_input",
        "_lazy_includes",
        "_lazy_include_cache",
        "_artifact_provenance",
        "name",
    ]
)


class FlowSpecMeta(type):
    def __new__(mcs, name, bases, namespace):
        cls = super().__new__(mcs, name, bases, namespace)
        if name == "FlowSpec":
            return cls
        # collect steps in definition order (MRO-merged: base steps first,
        # overridden steps keep the subclass version)
        steps = []
        params = []
        for klass in reversed(cls.__mro__):
            for attr_name, attr in vars(klass).items():
                if getattr(attr, "is_step", False):
                    if attr_name not in steps:
                        steps.append(attr_name)
                elif isinstance(attr, Parameter):
                    if attr_name not in [p[0] for p in params]:
                        params.append((attr_name, attr))
        cls._steps = steps
        cls._params = params
        return cls


class FlowSpec(object, metaclass=FlowSpecMeta):
    """Subclass this and write ``@step`` methods chained with
    ``self.next(...)`` to define a flow."""

    _flow_decorators = []

    def __init__(self, use_cli=True):
        self.name = type(self).__name__
        self._datastore = None
        self._foreach_stack = []
        self._transition = None
        self._parameter_names = [p[0] for p in type(self)._params]
        self._current_step = None
        self._cached_input = _NOT_SET
        if use_cli:
            from .cli import main

            main(self)

    # ------------------------------------------------------------------ graph
    @classmethod
    def _flow_graph(cls):
        from .graph import FlowGraph

        return FlowGraph(cls)

    # -------------------------------------------------------------- artifacts
    def __getattr__(self, name):
        # called only when normal lookup fails: try the task datastore
        if name.startswith("__") or name in INTERNAL_ATTRS:
            raise AttributeError(name)
        lazy = self.__dict__.get("_lazy_includes")
        if lazy and name in lazy:
            # IncludeFile parameter: decode from the CAS on first
            # access; cache OUTSIDE the artifact namespace so persist()
            # keeps propagating the small handle, not the content
            cache = self.__dict__.setdefault("_lazy_include_cache", {})
            if name not in cache:
                cache[name] = lazy[name].decode(
                    self.__dict__.get("_datastore"))
            return cache[name]
        ds = self.__dict__.get("_datastore")
        if ds is not None and name in ds:
            value = ds[name]
            # cache so repeated access doesn't re-deserialize
            object.__setattr__(self, name, value)
            # (provenance for persist()'s no-reserialize fast path is
            # recorded at deserialization time — task_datastore.py)
            return value
        raise AttributeError(
            "Flow %s has no artifact or attribute '%s'"
            % (self.__dict__.get("name", type(self).__name__), name)
        )

    def _artifacts_to_persist(self):
        """Names/values of instance attributes that are artifacts."""
        for k, v in self.__dict__.items():
            if k in INTERNAL_ATTRS or k.startswith("__"):
                continue
            yield k, v

    # ----------------------------------------------------------- foreach ctx
    @property
    def index(self):
        """Index of this task inside the innermost foreach, or None."""
        if self._foreach_stack:
            return self._foreach_stack[-1].index
        return None

    @property
    def input(self):
        """The element of the foreach list this task processes."""
        if self._cached_input is not _NOT_SET:
            return self._cached_input
        if not self._foreach_stack:
            return None
        frame = self._foreach_stack[-1]
        if frame.var is None:
            # num_parallel gang step: input is the node index
            value = frame.index
        else:
            seq = getattr(self, frame.var)
            from .unbounded_foreach import UnboundedForeachInput

            if isinstance(seq, UnboundedForeachInput):
                value = seq[frame.index]
            else:
                value = list(seq)[frame.index] if not hasattr(
                    seq, "__getitem__") else seq[frame.index]
        object.__setattr__(self, "_cached_input", value)
        return value

    def foreach_stack(self):
        """[(index, num_splits, input_value_or_None)] per nesting level.

        The third element is the resolved INPUT VALUE at that nesting
        level (reference flowspec.py:654-684) — the element of the
        foreach sequence this branch processes — not the variable name.
        None when the value cannot be resolved (e.g. the sequence
        artifact is unavailable, or a num_parallel gang frame).
        """
        out = []
        for frame in self._foreach_stack:
            value = None
            if frame.var is not None:
                try:
                    seq = getattr(self, frame.var)
                    from .unbounded_foreach import UnboundedForeachInput

                    if isinstance(seq, UnboundedForeachInput):
                        value = seq[frame.index]
                    else:
                        value = (seq[frame.index]
                                 if hasattr(seq, "__getitem__")
                                 else list(seq)[frame.index])
                except Exception:
                    value = None
            out.append((frame.index, frame.num_splits, value))
        return out

    # ------------------------------------------------------------ transitions
    def next(self, *dsts, **kwargs):
        """Record the transition out of the current step.

        Forms: self.next(self.a) | self.next(self.a, self.b) |
        self.next(self.a, foreach='var') |
        self.next(self.a, num_parallel=N) |
        self.next(self.a, self.b, condition='var')
        """
        step_name = self._current_step or "?"
        foreach = kwargs.pop("foreach", None)
        condition = kwargs.pop("condition", None)
        num_parallel = kwargs.pop("num_parallel", None)
        if kwargs:
            raise InvalidNextException(
                "Step %s: unknown self.next() keyword(s): %s"
                % (step_name, ", ".join(kwargs)))
        if self._transition is not None:
            raise InvalidNextException(
                "Step %s: self.next() called twice." % step_name)
        if not dsts:
            raise InvalidNextException(
                "Step %s: self.next() needs at least one target step."
                % step_name)

        funcs = []
        for dst in dsts:
            name = getattr(dst, "__name__", None)
            target = getattr(type(self), name, None) if name else None
            if name is None or target is None or not getattr(
                    target, "is_step", False):
                raise InvalidNextException(
                    "Step %s: self.next() targets must be steps of this "
                    "flow." % step_name)
            funcs.append(name)

        num_splits = None
        if foreach is not None:
            if len(funcs) != 1:
                raise InvalidNextException(
                    "Step %s: foreach takes exactly one target step."
                    % step_name)
            try:
                seq = getattr(self, foreach)
            except AttributeError:
                raise InvalidNextException(
                    "Step %s: foreach variable '%s' is not an artifact."
                    % (step_name, foreach))
            from .unbounded_foreach import UnboundedForeachInput

            if isinstance(seq, UnboundedForeachInput):
                num_splits = None  # unbounded: control task decides
            else:
                try:
                    num_splits = len(seq)
                except TypeError:
                    seq = list(seq)
                    setattr(self, foreach, seq)
                    num_splits = len(seq)
                if num_splits == 0:
                    raise InvalidNextException(
                        "Step %s: foreach over an empty sequence ('%s')."
                        % (step_name, foreach))
        elif num_parallel is not None:
            if len(funcs) != 1:
                raise InvalidNextException(
                    "Step %s: num_parallel takes exactly one target step."
                    % step_name)
            num_parallel = int(num_parallel)
            if num_parallel < 1:
                raise InvalidNextException(
                    "Step %s: num_parallel must be >= 1." % step_name)
            num_splits = num_parallel
        elif condition is not None:
            try:
                cond_val = getattr(self, condition)
            except AttributeError:
                raise InvalidNextException(
                    "Step %s: condition variable '%s' is not an artifact."
                    % (step_name, condition))
            if isinstance(cond_val, str) and cond_val in funcs:
                funcs = [cond_val]
            elif isinstance(cond_val, bool) and len(funcs) == 2:
                funcs = [funcs[0]] if cond_val else [funcs[1]]
            else:
                raise InvalidNextException(
                    "Step %s: condition '%s' must be a target step name or "
                    "a bool (with exactly two targets)."
                    % (step_name, condition))

        self._transition = {
            "out_funcs": funcs,
            "foreach": foreach,
            "condition": condition,
            "num_parallel": num_parallel,
            "num_splits": num_splits,
        }

    # ---------------------------------------------------------------- merging
    def merge_artifacts(self, inputs, exclude=None, include=None):
        """Merge artifacts from join inputs onto self.

        Artifacts that agree across all inputs are propagated; conflicting
        ones must be set on self (or excluded) before calling, else
        UnhandledInMergeArtifactsException is raised.
        """
        exclude = set(exclude or ())
        include = set(include or ())
        if exclude and include:
            raise MFXException(
                "merge_artifacts: exclude and include are mutually "
                "exclusive.")
        param_names = set(self._parameter_names or ())

        # name -> list of (input, sha-or-marker)
        seen = {}
        for inp in inputs:
            for name in inp._artifact_names():
                if name.startswith("_") or name == "name":
                    continue
                if name in param_names:
                    continue
                if name in exclude:
                    continue
                if include and name not in include:
                    continue
                seen.setdefault(name, []).append(inp)

        if include:
            missing = [n for n in include if n not in seen]
            if missing:
                raise MissingInMergeArtifactsException(
                    "merge_artifacts: artifact(s) %s not found in any "
                    "input." % ", ".join(missing), missing)

        unhandled = []
        for name, holders in seen.items():
            if name in self.__dict__:
                continue  # user already resolved it
            shas = {h._artifact_sha(name) for h in holders}
            if len(shas) == 1 and None not in shas:
                setattr(self, name, holders[0]._get_artifact(name))
            else:
                # fall back to value equality for un-sha'd inputs
                vals = [h._get_artifact(name) for h in holders]
                try:
                    all_eq = all(v == vals[0] for v in vals[1:])
                except Exception:
                    all_eq = False
                if all_eq:
                    setattr(self, name, vals[0])
                else:
                    unhandled.append(name)
        if unhandled:
            raise UnhandledInMergeArtifactsException(
                "merge_artifacts: artifact(s) %s have conflicting values "
                "in the inputs. Set them on self or pass exclude=[...]."
                % ", ".join(sorted(unhandled)), sorted(unhandled))

    # ------------------------------------------------------------------- misc
    def __iter__(self):
        """Iterate over step functions (definition order)."""
        return iter(getattr(self, s) for s in self._steps)

    def __str__(self):
        return "Flow(%s)" % self.name


class _NotSet(object):
    def __repr__(self):
        return "<not set>"


_NOT_SET = _NotSet()
