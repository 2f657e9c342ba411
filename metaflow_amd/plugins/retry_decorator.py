"""@retry: add scheduler-level retries to a step.

Parity target: /root/reference/metaflow/plugins/retry_decorator.py (40 LoC).
"""

from ..decorators import StepDecorator, make_step_decorator


class RetryDecorator(StepDecorator):
    name = "retry"
    defaults = {"times": 3, "minutes_between_retries": 0}

    def step_task_retry_count(self):
        return int(self.attributes["times"]), 0


retry = make_step_decorator(RetryDecorator)
