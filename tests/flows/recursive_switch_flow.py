"""Recursive switch: a step loops back to itself via a condition until
done (reference recursive_switch behavior)."""

import os

from metaflow_amd import FlowSpec, step


class RecursiveSwitchFlow(FlowSpec):
    @step
    def start(self):
        self.counter = 0
        self.next(self.work)

    @step
    def work(self):
        self.counter += 1
        self.route = "work" if self.counter < 3 else "finish"
        self.next(self.work, self.finish, condition="route")

    @step
    def finish(self):
        assert self.counter == 3, self.counter
        if os.environ.get("REC_FAIL") == "1":
            raise RuntimeError("planned failure at finish")
        self.total = self.counter * 10
        self.next(self.end)

    @step
    def end(self):
        assert self.total == 30


if __name__ == "__main__":
    RecursiveSwitchFlow()
