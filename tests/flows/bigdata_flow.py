import numpy as np

from metaflow_amd import FlowSpec, step


class BigDataFlow(FlowSpec):
    @step
    def start(self):
        # 64 MiB artifact -> exercises the raw-codec parallel-hash CAS path
        self.big = np.arange(8 << 20, dtype=np.int64)
        self.next(self.check)

    @step
    def check(self):
        assert self.big.shape == (8 << 20,)
        assert int(self.big[12345]) == 12345
        self.checksum = int(self.big[:: 1 << 18].sum())
        self.next(self.end)

    @step
    def end(self):
        assert self.checksum == int(
            np.arange(8 << 20, dtype=np.int64)[:: 1 << 18].sum())


if __name__ == "__main__":
    BigDataFlow()
