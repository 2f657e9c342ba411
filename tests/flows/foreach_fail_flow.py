"""Partial foreach failure: child with input==2 fails on the first run;
resume must clone the succeeded children and rerun only the failed one."""

import os

from metaflow_amd import FlowSpec, step


class ForeachFailFlow(FlowSpec):
    @step
    def start(self):
        self.items = [0, 1, 2, 3]
        self.next(self.work, foreach="items")

    @step
    def work(self):
        if os.environ.get("FF_FAIL") == "1" and self.input == 2:
            import time

            time.sleep(4)  # let the sibling children finish first
            raise RuntimeError("planned failure on item 2")
        self.val = self.input * 10
        self.next(self.join)

    @step
    def join(self, inputs):
        self.total = sum(i.val for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        assert self.total == 60, self.total


if __name__ == "__main__":
    ForeachFailFlow()
