"""@schedule flow decorator: cron metadata for deployment-time compilers.

Parity target: /root/reference/metaflow/plugins/aws/step_functions/
schedule_decorator.py — deployment-time only; stored on the flow class.
"""

from ..decorators import FlowDecorator, make_flow_decorator


class ScheduleDecorator(FlowDecorator):
    name = "schedule"
    defaults = {"cron": None, "hourly": False, "daily": True, "weekly": False}


schedule = make_flow_decorator(ScheduleDecorator)
