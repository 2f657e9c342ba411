from metaflow_amd import FlowSpec, Parameter, step


class LinearFlow(FlowSpec):
    alpha = Parameter("alpha", default=3, type=int)

    @step
    def start(self):
        self.x = 10
        self.msg = "hello"
        self.next(self.middle)

    @step
    def middle(self):
        self.x = self.x * self.alpha
        self.next(self.end)

    @step
    def end(self):
        self.final = self.x + 1
        assert self.msg == "hello"  # passdown across untouched step


if __name__ == "__main__":
    LinearFlow()
