from metaflow_amd import FlowSpec, Parameter, step


class SwitchFlow(FlowSpec):
    route = Parameter("route", default="fast", type=str)

    @step
    def start(self):
        self.choice = "fast_path" if self.route == "fast" else "slow_path"
        self.next(self.fast_path, self.slow_path, condition="choice")

    @step
    def fast_path(self):
        self.result = "fast"
        self.next(self.finish)

    @step
    def slow_path(self):
        self.result = "slow"
        self.next(self.finish)

    @step
    def finish(self):
        self.final = self.result
        self.next(self.end)

    @step
    def end(self):
        assert self.final in ("fast", "slow")


if __name__ == "__main__":
    SwitchFlow()
