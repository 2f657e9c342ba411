"""metaflow_amd — an MI355X-native ML workflow engine.

Re-implements the capabilities of Netflix/metaflow (FlowSpec/@step API,
AST-derived DAGs with foreach/join/switch, content-addressed artifact
datastore, local subprocess runtime with gang-scheduled @parallel steps,
resume/clone recovery, Flow/Run/Step/Task client) designed MI355X-first:
the gang scheduler pins ranks onto GPUs with RCCL over xGMI, and the
training hot path (metaflow_amd.ops) is hand-written CDNA4 HIP.
"""

from .flowspec import FlowSpec
from .decorators import step
from .parameters import Parameter, JSONType
from .includefile import IncludeFile
from .current import current
from .unbounded_foreach import UnboundedForeachInput
from .user_config import (
    Config,
    ConfigValue,
    FlowMutator,
    MutableFlow,
    StepMutator,
)
from .user_decorators import (
    USER_SKIP_STEP,
    UserStepDecorator,
    user_step_decorator,
)
from .plugins.retry_decorator import retry
from .plugins.catch_decorator import catch
from .plugins.timeout_decorator import timeout
from .plugins.resources_decorator import resources
from .plugins.environment_decorator import environment
from .plugins.parallel_decorator import parallel, torch_parallel
from .plugins.checkpoint_decorator import checkpoint
from .plugins.card_decorator import card
from .plugins.secrets_decorator import secrets
from .plugins.exit_hook_decorator import exit_hook
from .plugins.trigger_decorator import trigger, trigger_on_finish
from .plugins.project_decorator import project
from .plugins.schedule_decorator import schedule
from .client import (
    Metaflow,
    Flow,
    Run,
    Step,
    Task,
    DataArtifact,
    namespace,
    get_namespace,
    default_namespace,
    get_metadata,
    default_metadata,
)
from .multicore_utils import parallel_map, parallel_imap_unordered
from .profile_util import profile
from .datatools import ObjectStore
from .runner import (
    DeployedFlow,
    Deployer,
    NBDeployer,
    NBRunner,
    Runner,
    TriggeredRun,
)
from .speculative import speculative_generate

__version__ = "0.1.0"

S3 = None  # cloud datatools are not part of the single-node build (yet)

__all__ = [
    "FlowSpec",
    "step",
    "Parameter",
    "JSONType",
    "IncludeFile",
    "current",
    "UnboundedForeachInput",
    "Config",
    "ConfigValue",
    "FlowMutator",
    "StepMutator",
    "MutableFlow",
    "retry",
    "catch",
    "timeout",
    "resources",
    "environment",
    "parallel",
    "torch_parallel",
    "checkpoint",
    "card",
    "secrets",
    "exit_hook",
    "trigger",
    "trigger_on_finish",
    "project",
    "schedule",
    "Metaflow",
    "Flow",
    "Run",
    "Step",
    "Task",
    "DataArtifact",
    "namespace",
    "get_namespace",
    "default_namespace",
    "get_metadata",
    "default_metadata",
    "profile",
    "ObjectStore",
    "Runner",
    "Deployer",
    "DeployedFlow",
    "TriggeredRun",
    "NBRunner",
    "NBDeployer",
    "speculative_generate",
    "parallel_map",
    "parallel_imap_unordered",
]
