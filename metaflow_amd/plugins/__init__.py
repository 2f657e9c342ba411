"""Plugin registry.

Parity target: /root/reference/metaflow/plugins/__init__.py:11-195 —
declarative name -> class lists for step decorators so `--with name` and
static `@name` both resolve here.
"""

from .retry_decorator import RetryDecorator
from .catch_decorator import CatchDecorator
from .timeout_decorator import TimeoutDecorator
from .resources_decorator import ResourcesDecorator
from .environment_decorator import EnvironmentDecorator
from .parallel_decorator import ParallelDecorator, TorchParallelDecorator
from .checkpoint_decorator import CheckpointDecorator
from .project_decorator import ProjectDecorator
from .schedule_decorator import ScheduleDecorator
from .card_decorator import CardDecorator
from .secrets_decorator import SecretsDecorator
from .exit_hook_decorator import ExitHookDecorator
from .trigger_decorator import TriggerDecorator, TriggerOnFinishDecorator

STEP_DECORATORS = {
    cls.name: cls
    for cls in (
        RetryDecorator,
        CatchDecorator,
        TimeoutDecorator,
        ResourcesDecorator,
        EnvironmentDecorator,
        ParallelDecorator,
        TorchParallelDecorator,
        CheckpointDecorator,
        CardDecorator,
        SecretsDecorator,
    )
}

FLOW_DECORATORS = {
    cls.name: cls
    for cls in (ProjectDecorator, ScheduleDecorator, ExitHookDecorator,
                TriggerDecorator, TriggerOnFinishDecorator)
}

# merge metaflow_amd_extensions.* contributions (extension_support.py)
from ..extension_support import load_extensions as _load_ext

_load_ext(STEP_DECORATORS, FLOW_DECORATORS)
