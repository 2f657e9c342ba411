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
    defaults = {"backend": None, "context_parallel": 1,
                "tensor_parallel": 1, "pipeline_parallel": 1}

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
                from datetime import timedelta

                if backend == "nccl":
                    # bind this rank's device BEFORE init: RCCL derives
                    # the communicator device from the current device,
                    # and a late bind is the classic wedged-rendezvous
                    timeout_s = int(os.environ.get(
                        "MFX_RCCL_INIT_TIMEOUT", "300"))
                    torch.cuda.set_device(
                        int(os.environ.get("LOCAL_RANK", "0")))
                else:
                    timeout_s = int(os.environ.get(
                        "MFX_RCCL_INIT_TIMEOUT", "300"))
                try:
                    dist.init_process_group(
                        backend=backend,
                        timeout=timedelta(seconds=timeout_s))
                except Exception as e:
                    # fail LOUDLY and fast: the scheduler tears down the
                    # gang and retries it once on a fresh port
                    raise RuntimeError(
                        "torch.distributed init failed (backend=%s, "
                        "rank=%s/%s, addr=%s:%s): %s" % (
                            backend, os.environ.get("RANK"),
                            os.environ.get("WORLD_SIZE"),
                            os.environ.get("MASTER_ADDR"),
                            os.environ.get("MASTER_PORT"), e)) from e
            cp_attr = self.attributes.get("context_parallel") or 1
            cp = (dist.get_world_size() if cp_attr == "all"
                  else int(cp_attr))
            if cp > 1:
                self._make_grid(dist, cp, "cp")
            tp_attr = self.attributes.get("tensor_parallel") or 1
            tp = (dist.get_world_size() if tp_attr == "all"
                  else int(tp_attr))
            if tp > 1:
                self._make_grid(dist, tp, "tp")
            pp_attr = self.attributes.get("pipeline_parallel") or 1
            pp = (dist.get_world_size() if pp_attr == "all"
                  else int(pp_attr))
            if pp > 1:
                self._make_grid(dist, pp, "pp")
            try:
                return step_func(*args, **kwargs)
            finally:
                if dist.is_initialized():
                    dist.destroy_process_group()

        wrapped.__name__ = getattr(step_func, "__name__", "step")
        return wrapped

    def _make_grid(self, dist, inner, kind):
        """Partition the gang into a dp x inner grid (SURVEY §5:
        `@parallel(context_parallel=k)` / `tensor_parallel=k`):
        consecutive ranks form the inner group (ring-attention peers
        for cp — pass to LlamaForCausalLM/ring_attention — or
        Megatron-style shard peers for tp — pass to
        TPLlamaForCausalLM); same-index ranks across inner groups form
        the data-parallel gradient group (pass to
        FlatParamModel(group=...) for tp; for cp the WHOLE gang
        all-reduce stays correct because cp shards see the same batch).
        Exposed as ``current.parallel.<kind>_group`` / ``dp_group`` /
        ``<kind>_rank`` / ``dp_rank``."""
        from ..current import current
        from ..exceptions import MFXException

        world = dist.get_world_size()
        rank = dist.get_rank()
        if world % inner != 0:
            raise MFXException(
                "%s=%d must divide num_parallel=%d"
                % (kind, inner, world))
        in_group = dp_group = None
        # new_group must be called by ALL ranks for EVERY group
        for start in range(0, world, inner):
            g = dist.new_group(list(range(start, start + inner)))
            if start <= rank < start + inner:
                in_group = g
        for idx in range(inner):
            g = dist.new_group(list(range(idx, world, inner)))
            if rank % inner == idx:
                dp_group = g
        par = getattr(current, "parallel", None)
        info = dict(par._asdict()) if par is not None else {}
        info.update({
            "%s_group" % kind: in_group, "dp_group": dp_group,
            "%s_rank" % kind: rank % inner, "%s_degree" % kind: inner,
            "dp_rank": rank // inner, "dp_degree": world // inner,
        })
        from collections import namedtuple

        current._update_env(
            {"parallel": namedtuple("Parallel", info)(**info)})


parallel = make_step_decorator(ParallelDecorator)
torch_parallel = make_step_decorator(TorchParallelDecorator)
