"""@catch inside a foreach: the failed child is swallowed, its
FailureHandledByCatch artifact flows to the join (reference
catch_retry behavior)."""

from metaflow_amd import FlowSpec, catch, step


class CatchForeachFlow(FlowSpec):
    @step
    def start(self):
        self.items = [0, 1, 2]
        self.next(self.work, foreach="items")

    @catch(var="err")
    @step
    def work(self):
        if self.input == 1:
            raise ValueError("boom on 1")
        self.val = self.input * 10
        self.next(self.join)

    @step
    def join(self, inputs):
        self.total = sum(getattr(i, "val", 0) for i in inputs)
        self.failures = [str(i.err) for i in inputs
                         if getattr(i, "err", None)]
        self.next(self.end)

    @step
    def end(self):
        assert self.total == 20, self.total
        assert len(self.failures) == 1 and "boom" in self.failures[0]


if __name__ == "__main__":
    CatchForeachFlow()
