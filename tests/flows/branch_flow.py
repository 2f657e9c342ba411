from metaflow_amd import FlowSpec, step


class BranchFlow(FlowSpec):
    @step
    def start(self):
        self.common = 7
        self.next(self.a, self.b)

    @step
    def a(self):
        self.val = 1
        self.next(self.join)

    @step
    def b(self):
        self.val = 2
        self.next(self.join)

    @step
    def join(self, inputs):
        self.total = inputs.a.val + inputs.b.val
        self.merge_artifacts(inputs, exclude=["val"])
        assert self.common == 7
        self.next(self.end)

    @step
    def end(self):
        assert self.total == 3


if __name__ == "__main__":
    BranchFlow()
