"""Partial failure deep in a nested foreach: exactly one inner task
(outer=1, inner=1) fails on the first run; resume clones every other
task across both levels and reruns only the failed leaf + its joins."""

import os

from metaflow_amd import FlowSpec, step


class NestedForeachFailFlow(FlowSpec):
    @step
    def start(self):
        self.outer = [0, 1]
        self.next(self.mid, foreach="outer")

    @step
    def mid(self):
        self.o = self.input
        self.inner = [0, 1, 2]
        self.next(self.leaf, foreach="inner")

    @step
    def leaf(self):
        if os.environ.get("NF_FAIL") == "1" and self.o == 1 \
                and self.input == 1:
            import time

            time.sleep(4)  # let the sibling leaves finish first
            raise RuntimeError("planned failure at (1,1)")
        self.val = self.o * 100 + self.input
        self.next(self.join_i)

    @step
    def join_i(self, inputs):
        self.subtotal = sum(i.val for i in inputs)
        self.next(self.join_o)

    @step
    def join_o(self, inputs):
        self.total = sum(i.subtotal for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        # (0+1+2) + (100+101+102) = 306
        assert self.total == 306, self.total


if __name__ == "__main__":
    NestedForeachFailFlow()
