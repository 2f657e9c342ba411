from metaflow_amd import FlowSpec, UnboundedForeachInput, step


class ListUBF(UnboundedForeachInput):
    """UBF input backed by a list (the external-batch-system stand-in)."""

    def __init__(self, items):
        self.items = list(items)

    def __iter__(self):
        return iter(self.items)

    def __len__(self):
        return len(self.items)

    def __getitem__(self, i):
        if i is None:
            return self
        return self.items[i]


class UBFFlow(FlowSpec):
    @step
    def start(self):
        self.payload = ListUBF([3, 5, 7])
        self.next(self.work, foreach="payload")

    @step
    def work(self):
        self.doubled = self.input * 2
        self.next(self.join)

    @step
    def join(self, inputs):
        self.total = sum(i.doubled for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        assert self.total == 30, self.total


if __name__ == "__main__":
    UBFFlow()
