from metaflow_amd import FlowSpec, step


class ForeachFlow(FlowSpec):
    @step
    def start(self):
        self.items = [1, 2, 3, 4]
        self.next(self.work, foreach="items")

    @step
    def work(self):
        self.squared = self.input * self.input
        self.idx = self.index
        self.next(self.join)

    @step
    def join(self, inputs):
        self.total = sum(i.squared for i in inputs)
        self.indexes = sorted(i.idx for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        assert self.total == 1 + 4 + 9 + 16, self.total
        assert self.indexes == [0, 1, 2, 3]


if __name__ == "__main__":
    ForeachFlow()
