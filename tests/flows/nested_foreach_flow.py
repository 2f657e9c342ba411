from metaflow_amd import FlowSpec, step


class NestedForeachFlow(FlowSpec):
    @step
    def start(self):
        self.outer = [1, 2, 3]
        self.next(self.mid, foreach="outer")

    @step
    def mid(self):
        self.o = self.input
        self.inner = [10, 20]
        self.next(self.leaf, foreach="inner")

    @step
    def leaf(self):
        self.val = self.o * self.input
        self.next(self.inner_join)

    @step
    def inner_join(self, inputs):
        self.subtotal = sum(i.val for i in inputs)
        self.next(self.outer_join)

    @step
    def outer_join(self, inputs):
        self.total = sum(i.subtotal for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        # sum over o in {1,2,3} of o*(10+20) = 30*6 = 180
        assert self.total == 180, self.total


if __name__ == "__main__":
    NestedForeachFlow()
