from .deployer import DeployedFlow, Deployer, TriggeredRun
from .metaflow_runner import ExecutingRun, Runner

__all__ = ["Runner", "ExecutingRun", "Deployer", "DeployedFlow",
           "TriggeredRun"]
