"""Typed flow parameters exposed on the CLI and persisted per run.

Parity target: /root/reference/metaflow/parameters.py:276 (Parameter),
:89 (JSONTypeClass). Parameters become CLI options on `run`/`resume`, are
persisted once per run in the `_parameters` pseudo-task, and are readable
as attributes in every step.
"""

import builtins
import json

from .exceptions import ParameterException


class JSONTypeClass(object):
    """Marker type: the CLI value is parsed with json.loads."""

    name = "JSON"

    def convert(self, value):
        if isinstance(value, str):
            try:
                return json.loads(value)
            except json.JSONDecodeError as e:
                raise ParameterException("Invalid JSON: %s" % e)
        return value

    def __repr__(self):
        return "JSON"


JSONType = JSONTypeClass()


class Parameter(object):
    # class-level registry of definition order per flow class is kept by
    # FlowSpecMeta; Parameter itself is a plain descriptor-ish holder
    def __init__(
        self,
        name,
        default=None,
        type=None,
        help=None,
        required=False,
        show_default=True,
        separator=None,
        external_artifact=None,
    ):
        self.name = name
        self.default = default
        self.help = help
        self.required = required
        self.show_default = show_default
        self.separator = separator
        self.IS_PARAMETER = True
        if type is None and default is not None and not callable(default):
            type = builtins.type(default)
        self.type = type

    def convert(self, value):
        """Coerce a CLI string to the parameter's python type."""
        if value is None:
            return None
        if isinstance(self.type, JSONTypeClass):
            return self.type.convert(value)
        if self.type is bool:
            if isinstance(value, bool):
                return value
            return str(value).lower() in ("1", "true", "yes")
        if self.type in (int, float, str):
            return self.type(value)
        if self.separator and isinstance(value, str):
            return value.split(self.separator)
        return value

    def resolve_default(self, context=None):
        if callable(self.default) and not isinstance(self.default, type):
            return self.default(context)
        return self.default

    def __get__(self, obj, owner=None):
        """Non-data descriptor (reference task.py:191 read-only property
        exposure): instance attributes set by the task shadow this; when
        no instance value exists, an IncludeFile parameter decodes its
        CAS blob on demand, anything else falls through to FlowSpec's
        __getattr__ (datastore lookup) by raising AttributeError."""
        if obj is None:
            return self
        lazy = obj.__dict__.get("_lazy_includes")
        if lazy and self.name in lazy:
            cache = obj.__dict__.setdefault("_lazy_include_cache", {})
            if self.name not in cache:
                cache[self.name] = lazy[self.name].decode(
                    obj.__dict__.get("_datastore"))
            return cache[self.name]
        raise AttributeError(self.name)

    def __repr__(self):
        return "Parameter(%s)" % self.name
