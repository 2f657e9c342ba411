from .deployer import DeployedFlow, Deployer, TriggeredRun
from .metaflow_runner import ExecutingRun, Runner
from .nbrun import NBDeployer, NBRunner

__all__ = ["Runner", "ExecutingRun", "Deployer", "DeployedFlow",
           "TriggeredRun", "NBRunner", "NBDeployer"]
