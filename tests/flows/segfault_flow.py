import ctypes

from metaflow_amd import FlowSpec, step


class SegfaultFlow(FlowSpec):
    @step
    def start(self):
        ctypes.string_at(0)  # deliberate segfault
        self.next(self.end)

    @step
    def end(self):
        pass


if __name__ == "__main__":
    SegfaultFlow()
