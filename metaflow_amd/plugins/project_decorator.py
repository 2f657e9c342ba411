"""@project flow decorator: namespacing for production deployments.

Parity target: /root/reference/metaflow/plugins/project_decorator.py —
computes the (project, branch) namespace, exposes
``current.project_name`` / ``branch_name`` / ``project_flow_name`` /
``is_production``, and tags every run with ``project:`` /
``project_branch:`` so the Client can filter deployments.
"""

import getpass
import os

from ..decorators import FlowDecorator, make_flow_decorator


class ProjectDecorator(FlowDecorator):
    name = "project"
    defaults = {"name": None, "branch": None, "production": False}

    def flow_init(self, flow_cls, graph, environment, flow_datastore,
                  metadata, logger, echo, options):
        from ..current import current

        pname = self.attributes.get("name") or flow_cls.__name__.lower()
        production = bool(self.attributes.get("production"))
        if production:
            branch = self.attributes.get("branch") or "prod"
        else:
            user = os.environ.get("USER")
            if not user:
                try:
                    user = getpass.getuser()
                except Exception:
                    user = "unknown"
            branch = self.attributes.get("branch") or "user.%s" % user
        current._update_env({
            "project_name": pname,
            "branch_name": branch,
            "is_production": production,
            "project_flow_name": "%s.%s.%s" % (pname, branch,
                                               flow_cls.__name__),
        })
        flow_cls._project_tags = ["project:%s" % pname,
                                  "project_branch:%s" % branch]


project = make_flow_decorator(ProjectDecorator)
