"""@environment: inject env vars into the task process.

Parity target: /root/reference/metaflow/plugins/environment_decorator.py.
"""

from ..decorators import StepDecorator, make_step_decorator


class EnvironmentDecorator(StepDecorator):
    name = "environment"
    defaults = {"vars": {}}

    def runtime_step_cli(self, args, retry_count, max_user_code_retries,
                         ubf_context):
        args["env"].update(
            {str(k): str(v) for k, v in self.attributes["vars"].items()})


environment = make_step_decorator(EnvironmentDecorator)
