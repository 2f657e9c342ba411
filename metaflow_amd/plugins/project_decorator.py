"""@project flow decorator: namespacing for production deployments.

Parity target: /root/reference/metaflow/plugins/project_decorator.py.
"""

from ..decorators import FlowDecorator, make_flow_decorator


class ProjectDecorator(FlowDecorator):
    name = "project"
    defaults = {"name": None, "branch": None, "production": False}


project = make_flow_decorator(ProjectDecorator)
