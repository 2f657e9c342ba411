from metaflow_amd import FlowSpec, catch, step


class CatchFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.will_fail)

    @catch(var="failure")
    @step
    def will_fail(self):
        raise ValueError("intentional")
        self.next(self.end)  # noqa: unreachable on purpose

    @step
    def end(self):
        assert self.failure is not None
        assert "intentional" in self.failure.exception


if __name__ == "__main__":
    CatchFlow()
