"""@resources: declarative cpu/gpu/memory attributes.

Parity target: /root/reference/metaflow/plugins/resources_decorator.py
(44 LoC). On a single MI355X node this drives GPU-count reservation for the
gang scheduler; it is also consumed by @parallel for rank->GPU pinning.
"""

from ..decorators import StepDecorator, make_step_decorator


class ResourcesDecorator(StepDecorator):
    name = "resources"
    defaults = {"cpu": 1, "gpu": 0, "memory": 4096, "shared_memory": None}


resources = make_step_decorator(ResourcesDecorator)
