from metaflow_amd import FlowSpec, current, parallel, step


class ParallelFlow(FlowSpec):
    @step
    def start(self):
        self.n = 4
        self.next(self.train, num_parallel=4)

    @parallel
    @step
    def train(self):
        assert current.parallel.num_nodes == 4
        self.node_index = current.parallel.node_index
        self.main_ip = current.parallel.main_ip
        self.control = current.parallel.control_task_id
        self.next(self.join)

    @step
    def join(self, inputs):
        indexes = sorted(i.node_index for i in inputs)
        assert indexes == [0, 1, 2, 3], indexes
        assert len({i.control for i in inputs}) == 1
        self.ok = True
        self.next(self.end)

    @step
    def end(self):
        assert self.ok


if __name__ == "__main__":
    ParallelFlow()
