"""Unbounded-foreach support.

Parity target: /root/reference/metaflow/unbounded_foreach.py:6 and the
control/mapper protocol in plugins/test_unbounded_foreach_decorator.py.

A foreach over an UnboundedForeachInput schedules a single *control* task;
the control task materializes/launches the mappers itself and persists
``_control_mapper_tasks`` (a list of mapper pathspecs) so the join can gate
on them.
"""

UBF_CONTROL = "ubf_control"
UBF_TASK = "ubf_task"

CONTROL_TASK_TAG = "control_task"


class UnboundedForeachInput(object):
    """Marker base class: foreach over this yields a control task instead of
    a static fan-out."""

    NAME = "UnboundedForeachInput"

    def __getitem__(self, item):
        # control task: index is None → return self; mappers index normally
        if item is None:
            return self
        raise NotImplementedError(
            "Subclasses of UnboundedForeachInput that support indexing "
            "must implement __getitem__")

    def __repr__(self):
        return self.NAME
