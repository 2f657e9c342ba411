import os

from metaflow_amd import FlowSpec, current, retry, step


class RetryFlow(FlowSpec):
    @step
    def start(self):
        self.marker_dir = os.environ["RETRY_MARKER_DIR"]
        self.next(self.flaky)

    @retry(times=2)
    @step
    def flaky(self):
        marker = os.path.join(self.marker_dir, "attempt_%d"
                              % current.retry_count)
        open(marker, "w").close()
        if current.retry_count < 2:
            raise RuntimeError("flaky failure on attempt %d"
                               % current.retry_count)
        self.attempts_seen = current.retry_count
        self.next(self.end)

    @step
    def end(self):
        assert self.attempts_seen == 2


if __name__ == "__main__":
    RetryFlow()
