"""@parallel / @torch_parallel: gang-step support inside the task process.

Parity target: /root/reference/metaflow/plugins/parallel_decorator.py and
plugins/frameworks/pytorch.py:27-46. The MI355X design moves gang LAUNCH into
the scheduler (runtime._queue_gang: N rank processes, MFX_PARALLEL_* env,
HIP_VISIBLE_DEVICES pinning); these decorators handle the in-task side:
exposing current.parallel and initializing torch.distributed over RCCL
(backend "nccl" IS RCCL on ROCm) via env:// rendezvous on 127.0.0.1.
"""

import os

from ..decorators import StepDecorator, make_step_decorator


class ParallelDecorator(StepDecorator):
    name = "parallel"
    defaults = {}

    def task_pre_step(self, step_name, task_datastore, metadata, run_id,
                      task_id, flow, graph, retry_count,
                      max_user_code_retries, ubf_context, inputs):
        # current.parallel is populated by task.py from MFX_PARALLEL_* env;
        # nothing else to do for the plain gang decorator
        pass


class TorchParallelDecorator(ParallelDecorator):
    """Initialize (and tear down) a torch.distributed process group around
    the user step. One process per GPU, RCCL over xGMI when CUDA/HIP devices
    are visible, gloo otherwise (CPU tests)."""

    name = "torch_parallel"
    defaults = {"backend": None}

    def task_decorate(self, step_func, flow, graph, retry_count,
                      max_user_code_retries, ubf_context):
        if not os.environ.get("MFX_PARALLEL_NUM_NODES"):
            return step_func

        def wrapped(*args, **kwargs):
            import torch
            import torch.distributed as dist

            backend = self.attributes.get("backend")
            if backend is None:
                num_nodes = int(os.environ.get("MFX_PARALLEL_NUM_NODES",
                                               "1"))
                n_gpus = (torch.cuda.device_count()
                          if torch.cuda.is_available() else 0)
                # RCCL needs one DISTINCT device per rank; oversubscribed
                # gangs (e.g. CPU tests or a 2-rank gang on a 1-GPU box)
                # fall back to gloo. The scheduler pins one visible device
                # per rank, so n_gpus is what THIS rank can see.
                visible = int(os.environ.get("MFX_PARALLEL_TOTAL_GPUS",
                                             str(n_gpus)))
                backend = "nccl" if (n_gpus > 0 and num_nodes <= visible) \
                    else "gloo"
            if not dist.is_initialized():
                dist.init_process_group(backend=backend)
            try:
                return step_func(*args, **kwargs)
            finally:
                if dist.is_initialized():
                    dist.destroy_process_group()

        wrapped.__name__ = getattr(step_func, "__name__", "step")
        return wrapped


parallel = make_step_decorator(ParallelDecorator)
torch_parallel = make_step_decorator(TorchParallelDecorator)
