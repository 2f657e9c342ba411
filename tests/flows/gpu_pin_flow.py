"""GPU pinning: with 4 (fake) GPUs and @resources(gpu=2), a 2-rank gang
must see disjoint HIP_VISIBLE_DEVICES pairs."""

import os

from metaflow_amd import FlowSpec, current, resources, step, torch_parallel


class GpuPinFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.work, num_parallel=2)

    @resources(gpu=2)
    @torch_parallel
    @step
    def work(self):
        self.devs = os.environ.get("HIP_VISIBLE_DEVICES")
        self.rank = current.parallel.node_index
        self.next(self.join)

    @step
    def join(self, inputs):
        pins = {i.rank: i.devs for i in inputs}
        assert pins[0] == "0,1" and pins[1] == "2,3", pins
        self.pins = pins
        self.next(self.end)

    @step
    def end(self):
        pass


if __name__ == "__main__":
    GpuPinFlow()
