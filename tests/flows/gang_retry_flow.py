"""Gang transient-failure recovery: rank 1 dies on the FIRST attempt
only (marker file); the runtime must tear the gang down and retry it as
a unit on a fresh rendezvous port, and the retry completes."""

import os

from metaflow_amd import FlowSpec, current, step, torch_parallel


class GangRetryFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.work, num_parallel=2)

    @torch_parallel
    @step
    def work(self):
        import torch.distributed as dist

        rank = current.parallel.node_index
        marker = os.path.join(os.environ["GANG_RETRY_DIR"],
                              "attempted_%d" % rank)
        first_attempt = not os.path.exists(marker)
        with open(marker, "a") as f:
            f.write("x")
        if first_attempt and rank == 1:
            os._exit(3)  # simulated rendezvous-class crash
        import torch

        t = torch.tensor([float(rank)])
        dist.all_reduce(t)
        self.gang_sum = float(t)
        self.rank = rank
        self.port = os.environ.get("MFX_PARALLEL_MAIN_PORT")
        self.next(self.join)

    @step
    def join(self, inputs):
        self.ranks = sorted(i.rank for i in inputs)
        self.gang_sum = inputs[0].gang_sum
        self.next(self.end)

    @step
    def end(self):
        assert self.ranks == [0, 1], self.ranks
        assert self.gang_sum == 1.0


if __name__ == "__main__":
    GangRetryFlow()
