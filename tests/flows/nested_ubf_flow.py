"""Unbounded foreach nested inside a static foreach (reference
nested_unbounded_foreach behavior)."""

from metaflow_amd import FlowSpec, UnboundedForeachInput, step


class ListUBF(UnboundedForeachInput):
    def __init__(self, items):
        self.items = list(items)

    def __iter__(self):
        return iter(self.items)

    def __len__(self):
        return len(self.items)

    def __getitem__(self, i):
        return self if i is None else self.items[i]


class NestedUBFFlow(FlowSpec):
    @step
    def start(self):
        self.outer = [10, 20]
        self.next(self.mid, foreach="outer")

    @step
    def mid(self):
        self.base = self.input
        self.payload = ListUBF(range(3))
        self.next(self.work, foreach="payload")

    @step
    def work(self):
        self.val = self.base + self.input
        self.next(self.join_u)

    @step
    def join_u(self, inputs):
        self.subtotal = sum(i.val for i in inputs)
        self.next(self.join_o)

    @step
    def join_o(self, inputs):
        self.total = sum(i.subtotal for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        # (10+0)+(10+1)+(10+2) + (20+0)+(20+1)+(20+2) = 33 + 63 = 96
        assert self.total == 96, self.total


if __name__ == "__main__":
    NestedUBFFlow()
