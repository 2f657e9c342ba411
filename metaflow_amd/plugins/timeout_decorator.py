"""@timeout: SIGALRM-based step timeout.

Parity target: /root/reference/metaflow/plugins/timeout_decorator.py (106 LoC).
"""

import signal

from ..decorators import StepDecorator, make_step_decorator
from ..exceptions import MFXException


class TimeoutException(MFXException):
    headline = "Step timed out"


class TimeoutDecorator(StepDecorator):
    name = "timeout"
    defaults = {"seconds": 0, "minutes": 0, "hours": 0}

    def _secs(self):
        return (int(self.attributes["seconds"])
                + 60 * int(self.attributes["minutes"])
                + 3600 * int(self.attributes["hours"]))

    def task_decorate(self, step_func, flow, graph, retry_count,
                      max_user_code_retries, ubf_context):
        secs = self._secs()
        if secs <= 0:
            return step_func

        def timed(*args, **kwargs):
            def handler(signum, frame):
                raise TimeoutException(
                    "Step exceeded its timeout of %d seconds." % secs)

            old = signal.signal(signal.SIGALRM, handler)
            signal.alarm(secs)
            try:
                return step_func(*args, **kwargs)
            finally:
                signal.alarm(0)
                signal.signal(signal.SIGALRM, old)

        timed.__name__ = getattr(step_func, "__name__", "step")
        return timed


timeout = make_step_decorator(TimeoutDecorator)
