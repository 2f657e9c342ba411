"""@checkpoint: sharded model/optimizer checkpointing through the CAS.

The reference has no @checkpoint (it lives in an external
extension); this is a new capability required by BASELINE config 4. Each
rank writes its own shard blobs (content-hashed -> unchanged shards dedup
to no-ops); an index artifact maps shard names to CAS keys so resume and the
Client see checkpoints like any artifact. The fast GPU path (pinned-buffer
hipMemcpyAsync staging on a side stream) lives in
metaflow_amd/parallel/checkpoint.py; this decorator wires `current.checkpoint`.
"""

from ..decorators import StepDecorator, make_step_decorator


class Checkpointer(object):
    def __init__(self, flow, task_datastore):
        self._flow = flow
        self._ds = task_datastore

    def save(self, state_dict, name="checkpoint"):
        """Persist a (possibly GPU-resident) state dict as per-tensor CAS
        blobs + an index artifact. Returns the index."""
        from ..parallel.checkpoint import save_state_dict

        return save_state_dict(self._ds, state_dict, name)

    def load(self, name="checkpoint", map_location="cpu", source=None):
        from ..parallel.checkpoint import load_state_dict

        ds = source if source is not None else self._ds
        return load_state_dict(ds, name, map_location)


class CheckpointDecorator(StepDecorator):
    name = "checkpoint"
    defaults = {}

    def task_pre_step(self, step_name, task_datastore, metadata, run_id,
                      task_id, flow, graph, retry_count,
                      max_user_code_retries, ubf_context, inputs):
        from ..current import current

        current._update_env(
            {"checkpoint": Checkpointer(flow, task_datastore)})


checkpoint = make_step_decorator(CheckpointDecorator)
