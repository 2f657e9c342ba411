from metaflow_amd import FlowSpec, IncludeFile, step


class IncludeFlow(FlowSpec):
    data_file = IncludeFile("data_file", required=True)

    @step
    def start(self):
        self.n_lines = len(self.data_file.splitlines())
        self.next(self.end)

    @step
    def end(self):
        assert self.n_lines == 3, self.n_lines


if __name__ == "__main__":
    IncludeFlow()
