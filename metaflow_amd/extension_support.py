"""Extension-package discovery.

Parity target: /root/reference/metaflow/extension_support/__init__.py
(`_get_extension_packages`: discovers ``metaflow_extensions.*`` namespace
packages and merges their plugin contributions). Here the namespace is
``metaflow_amd_extensions``: any installed package (or any directory on
``PYTHONPATH`` containing ``metaflow_amd_extensions/<name>/__init__.py``)
is imported at plugin-registry load and may contribute step/flow
decorators.

An extension module contributes either of:

* ``STEP_DECORATORS`` / ``FLOW_DECORATORS`` — dicts (or lists of classes
  with ``.name``) merged into the registry;
* ``get_plugins()`` returning ``{"step_decorators": ..., "flow_decorators":
  ...}``.

A broken extension logs a warning and is skipped — extensions must never
take down the engine (reference behavior).
"""

import importlib
import pkgutil
import sys

EXT_NAMESPACE = "metaflow_amd_extensions"


def _as_dict(contrib):
    if contrib is None:
        return {}
    if isinstance(contrib, dict):
        return dict(contrib)
    return {cls.name: cls for cls in contrib}


def iter_extension_modules():
    """Yield imported extension submodules under the namespace package."""
    try:
        ns = importlib.import_module(EXT_NAMESPACE)
    except ImportError:
        return
    paths = list(getattr(ns, "__path__", []))
    if not paths:
        return
    for info in pkgutil.iter_modules(paths):
        name = "%s.%s" % (EXT_NAMESPACE, info.name)
        try:
            yield importlib.import_module(name)
        except Exception as ex:  # noqa: BLE001 — never kill the engine
            sys.stderr.write(
                "[mfx] warning: extension %s failed to load: %r\n"
                % (name, ex))


def load_extensions(step_decorators, flow_decorators):
    """Merge every discovered extension's contributions into the given
    registries (in-place; extensions override built-ins by name, last
    one wins in module-name order)."""
    for mod in iter_extension_modules():
        contribs = {}
        get_plugins = getattr(mod, "get_plugins", None)
        if callable(get_plugins):
            try:
                contribs = get_plugins() or {}
            except Exception as ex:  # noqa: BLE001
                sys.stderr.write(
                    "[mfx] warning: %s.get_plugins() failed: %r\n"
                    % (mod.__name__, ex))
                continue
        step_decorators.update(_as_dict(
            contribs.get("step_decorators",
                         getattr(mod, "STEP_DECORATORS", None))))
        flow_decorators.update(_as_dict(
            contribs.get("flow_decorators",
                         getattr(mod, "FLOW_DECORATORS", None))))
