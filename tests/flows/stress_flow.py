"""Scheduler stress: wide foreach under the worker cap with random
transient task failures recovered by @retry."""

import os
import random

from metaflow_amd import FlowSpec, current, retry, step


class StressFlow(FlowSpec):
    @step
    def start(self):
        self.items = list(range(48))
        self.next(self.work, foreach="items")

    @retry(times=3)
    @step
    def work(self):
        # ~25% of first attempts fail; retries must recover all of them
        rng = random.Random(self.input)
        if current.retry_count == 0 and rng.random() < 0.25:
            raise RuntimeError("transient failure on item %d" % self.input)
        self.value = self.input * 2
        self.attempt_used = current.retry_count
        self.next(self.join)

    @step
    def join(self, inputs):
        vals = sorted(i.value for i in inputs)
        assert vals == [i * 2 for i in range(48)], vals
        self.retried = sum(1 for i in inputs if i.attempt_used > 0)
        self.next(self.end)

    @step
    def end(self):
        assert self.retried > 0, "expected at least one retried task"


if __name__ == "__main__":
    StressFlow()
