"""@catch: swallow a step's exception after retries are exhausted and record
it in an artifact instead of failing the run.

Parity target: /root/reference/metaflow/plugins/catch_decorator.py (121 LoC):
task_exception returning True + a fallback artifact.
"""

from ..decorators import StepDecorator, make_step_decorator


class FailureHandledByCatch(object):
    def __init__(self, exception_repr):
        self.exception = exception_repr

    def __repr__(self):
        return "FailureHandledByCatch(%s)" % self.exception

    def __bool__(self):
        # truthy so `if self.failed:` works naturally
        return True


class CatchDecorator(StepDecorator):
    name = "catch"
    defaults = {"var": None, "print_exception": True}

    def step_init(self, flow, graph, step_name, decorators, datastore_type,
                  logger):
        # A swallowed exception means the step never called self.next(),
        # so the runtime must synthesize the transition from the static
        # graph. That is only well-defined for linear/static-split/join
        # shapes: a foreach split has no split size, a switch picks one
        # branch at runtime, and a parallel split has no gang width.
        # The reference likewise rejects @catch on foreach splits
        # (reference catch_decorator.py:45-52).
        node = graph[step_name]
        if node.type in ("foreach", "split-switch", "split-parallel"):
            from ..exceptions import GraphException

            raise GraphException(
                "@catch is not supported on step '%s': a %s step that "
                "fails cannot choose its transition (the split size / "
                "branch / gang width is computed inside the step). Move "
                "@catch into the child steps instead."
                % (step_name, node.type))

    def task_exception(self, exception, step_name, flow, graph, retry_count,
                       max_user_code_retries):
        # only swallow on the final attempt; earlier attempts should retry
        if retry_count < max_user_code_retries:
            return False
        var = self.attributes.get("var")
        if var:
            setattr(flow, var, FailureHandledByCatch(repr(exception)))
        if self.attributes.get("print_exception"):
            import traceback

            traceback.print_exc()
        # @catch swallows the exception, but the step never called
        # self.next(); synthesize the transition so the flow continues
        if flow._transition is None:
            node = graph[step_name]
            if node.out_funcs:
                flow._transition = {
                    "out_funcs": list(node.out_funcs),
                    "foreach": None,
                    "condition": None,
                    "num_parallel": None,
                    "num_splits": None,
                }
        return True


catch = make_step_decorator(CatchDecorator)
