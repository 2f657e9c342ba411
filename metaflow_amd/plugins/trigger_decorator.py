"""@trigger / @trigger_on_finish flow decorators.

Parity target: /root/reference/metaflow/plugins/events_decorator.py.
On a single node there is no event bus; the trigger spec is recorded in
run metadata so deployment-time compilers (or a future event daemon) can
consume it, and `current.trigger` is populated when a run is started with
MFX_TRIGGER_EVENT set (JSON payload).
"""

import json
import os

from ..decorators import FlowDecorator, make_flow_decorator


class TriggerInfo(object):
    def __init__(self, name=None, payload=None, run_pathspec=None):
        self.name = name
        self.payload = payload or {}
        self.run_pathspec = run_pathspec

    def __repr__(self):
        return "TriggerInfo(%s)" % self.name

    @classmethod
    def from_env(cls):
        raw = os.environ.get("MFX_TRIGGER_EVENT")
        if not raw:
            return None
        try:
            data = json.loads(raw)
        except ValueError:
            data = {"name": raw}
        return cls(name=data.get("name"), payload=data.get("payload"),
                   run_pathspec=data.get("run"))


class TriggerDecorator(FlowDecorator):
    name = "trigger"
    defaults = {"event": None, "events": []}


class TriggerOnFinishDecorator(FlowDecorator):
    name = "trigger_on_finish"
    defaults = {"flow": None, "flows": []}


trigger = make_flow_decorator(TriggerDecorator)
trigger_on_finish = make_flow_decorator(TriggerOnFinishDecorator)
