from .metaflow_runner import Runner, ExecutingRun

__all__ = ["Runner", "ExecutingRun"]
