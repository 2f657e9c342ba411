from metaflow_amd import FlowSpec, current, project, step


@project(name="mlplat")
class ProjectFlow(FlowSpec):
    @step
    def start(self):
        self.pname = current.project_name
        self.branch = current.branch_name
        self.pflow = current.project_flow_name
        assert not current.is_production
        self.next(self.end)

    @step
    def end(self):
        pass


if __name__ == "__main__":
    ProjectFlow()
