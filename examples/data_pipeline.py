"""Example: a fan-out data pipeline with retries, catch, and bulk IO.

    python examples/data_pipeline.py run --shards 16
"""

import numpy as np

from metaflow_amd import FlowSpec, ObjectStore, Parameter, catch, retry, step


class DataPipeline(FlowSpec):
    shards = Parameter("shards", default=16, type=int)

    @step
    def start(self):
        self.shard_ids = list(range(int(self.shards)))
        self.next(self.process, foreach="shard_ids")

    @retry(times=2)
    @catch(var="failure")
    @step
    def process(self):
        # synthetic shard work: decode -> transform -> stats
        rng = np.random.default_rng(self.input)
        data = rng.standard_normal(1 << 18).astype(np.float32)
        self.mean = float(data.mean())
        self.std = float(data.std())
        self.shard = self.input
        self.next(self.merge)

    @step
    def merge(self, inputs):
        ok = [i for i in inputs if not getattr(i, "failure", None)]
        self.stats = {i.shard: (i.mean, i.std) for i in ok}
        self.n_ok = len(ok)
        self.next(self.publish)

    @step
    def publish(self):
        import json
        import tempfile

        out = tempfile.mkdtemp(prefix="pipeline_out_")
        with ObjectStore(out) as store:
            store.put("stats.json",
                      json.dumps(self.stats).encode())
        self.published_to = out
        self.next(self.end)

    @step
    def end(self):
        assert self.n_ok == int(self.shards)
        print("published to", self.published_to)


if __name__ == "__main__":
    DataPipeline()
