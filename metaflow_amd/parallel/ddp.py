"""Flat-buffer data parallelism over RCCL/xGMI + fused AdamW.

MI355X-first design (SURVEY §5 'distributed communication backend'):

* all bf16 params live in ONE contiguous buffer (params become views), and
  all grads in a matching flat buffer — autograd accumulates straight into
  the flat slices (no per-tensor .grad allocations, no copy into buckets);
* the flat grad buffer is divided into contiguous buckets (~64 MiB: sized
  for xGMI per-link ring bandwidth, not NVSwitch); as backward finishes the
  last param of a bucket, an async all-reduce(AVG) is launched so
  communication overlaps the rest of backward;
* the optimizer is ONE fused AdamW kernel call over the whole flat buffer
  (ops/csrc/adam.hip) — fp32 m/v (+optional fp32 master weights), so a
  full 8B update is a handful of kernel launches instead of ~300.
"""

import torch
import torch.distributed as dist

from ..ops import kernels as K


class FlatParamModel(object):
    """Flattens a module's parameters into one bf16 buffer + flat grads."""

    def __init__(self, module, bucket_mb=64, group=None, zero=False):
        """group: the process group to average gradients over (default
        WORLD). For dp x tp grids pass the DP group — sharded (tp)
        params must only all-reduce across ranks holding the same
        shard.

        zero=True enables ZeRO-1 optimizer-state sharding: gradients
        reduce-scatter (each rank keeps ONE contiguous shard of the
        averaged flat grad), FusedAdamW holds fp32 m/v (+master) for
        that shard only, and updated bf16 params all-gather after the
        step. Cuts optimizer memory by the DP world size — what lets a
        70B (840 GB of replicated train state) fit 8 x 288 GB. Bucketed
        backward overlap is disabled in this mode (one reduce-scatter
        at finish_grad_sync); prefer plain DDP when memory allows."""
        self.group = group
        self.zero = bool(zero)
        self.module = module
        all_params = [p for p in module.parameters() if p.requires_grad]
        # params marked _mfx_no_sync (e.g. expert-parallel weights whose
        # grads are already complete after the token all-to-all) are placed
        # AFTER the sync region so buckets never cover them
        sync = [p for p in all_params
                if not getattr(p, "_mfx_no_sync", False)]
        nosync = [p for p in all_params
                  if getattr(p, "_mfx_no_sync", False)]
        params = sync + nosync
        self.n_sync_params = len(sync)
        self.params = params
        self.zero_world = 1
        if self.zero and dist.is_available() and dist.is_initialized():
            self.zero_world = dist.get_world_size(self.group)
        # ZeRO-1 + expert parallelism: only the SYNC region is sharded
        # (reduce-scattered / all-gathered); _mfx_no_sync (expert)
        # params differ per rank — their grads are already complete
        # after the token all-to-all and their optimizer state stays
        # owner-local and full. The buffer is laid out
        # [sync | align-gap | nosync] so the sync region splits into
        # world equal shards (each a multiple of 4 for fused adam).
        sync_sum = sum(self._padded(p.numel()) for p in sync)
        if self.zero_world > 1:
            align = 4 * self.zero_world
            self.sync_total = (sync_sum + align - 1) // align * align
        else:
            self.sync_total = sync_sum
        total = self.sync_total + sum(self._padded(p.numel())
                                      for p in nosync)
        self.total = total
        device = params[0].device
        dtype = params[0].dtype
        self.flat_param = torch.empty(total, dtype=dtype, device=device)
        self.flat_grad = torch.zeros(total, dtype=dtype, device=device)

        offset = 0
        self.offsets = []
        for i, p in enumerate(params):
            if i == self.n_sync_params:
                offset = self.sync_total  # skip the alignment gap
            n = p.numel()
            self.flat_param[offset:offset + n].copy_(p.detach().reshape(-1))
            # re-point the parameter at the flat storage
            p.data = self.flat_param[offset:offset + n].view_as(p)
            p.grad = self.flat_grad[offset:offset + n].view_as(p)
            self.offsets.append(offset)
            offset += self._padded(n)

        # buckets: contiguous ranges of the SYNC region of the flat buffer
        bucket_elems = (bucket_mb << 20) // self.flat_param.element_size()
        self.buckets = []  # (start, end, last_param_index)
        start = 0
        for i, p in enumerate(params[:self.n_sync_params]):
            end = self.offsets[i] + self._padded(p.numel())
            if end - start >= bucket_elems or i == self.n_sync_params - 1:
                self.buckets.append([start, end, i])
                start = end
        self.sync_end = self.buckets[-1][1] if self.buckets else 0
        self._pending = []
        self._hooks = []
        if self.zero_world > 1:
            ss = self.sync_total // self.zero_world
            r = dist.get_rank(self.group)
            self.zero_shard = (r * ss, (r + 1) * ss)
        else:
            self.zero_shard = (0, total)
        # optimizer segment OUTSIDE the sharded region (owner-local
        # expert params under ZeRO; empty otherwise — the plain path's
        # zero_shard already spans the whole buffer)
        self.local_seg = (self.sync_total, total) \
            if self.zero_world > 1 and nosync else None

    @staticmethod
    def _padded(n):
        return (n + 3) & ~3  # fused adam wants numel % 4 == 0

    def zero_grad(self):
        self.flat_grad.zero_()

    # ----------------------------------------------------------- allreduce
    def install_overlap_hooks(self):
        """Launch each bucket's all-reduce as soon as its last param's grad
        is accumulated (params complete roughly in reverse order, so buckets
        are checked by completion count)."""
        if not (dist.is_available() and dist.is_initialized()
                and dist.get_world_size(self.group) > 1):
            return
        if self.zero:
            return  # ZeRO-1: one reduce-scatter in finish_grad_sync
        self._done = set()
        bucket_last_param = {}
        prev = -1
        for bi, (_s, _e, last) in enumerate(self.buckets):
            for pi in range(prev + 1, last + 1):
                bucket_last_param[pi] = None
            bucket_last_param[last] = bi
            prev = last
        bucket_sizes = {}
        prev = -1
        for bi, (_s, _e, last) in enumerate(self.buckets):
            bucket_sizes[bi] = set(range(prev + 1, last + 1))
            prev = last
        self._bucket_param_sets = bucket_sizes
        self._bucket_done_count = {bi: 0 for bi in range(len(self.buckets))}
        param_to_bucket = {}
        prev = -1
        for bi, (_s, _e, last) in enumerate(self.buckets):
            for pi in range(prev + 1, last + 1):
                param_to_bucket[pi] = bi
            prev = last

        for pi, p in enumerate(self.params[:self.n_sync_params]):
            bi = param_to_bucket[pi]

            def hook(_param, bi=bi):
                self._bucket_done_count[bi] += 1
                if self._bucket_done_count[bi] == len(
                        self._bucket_param_sets[bi]):
                    s, e, _ = self.buckets[bi]
                    work = dist.all_reduce(self.flat_grad[s:e],
                                           group=self.group,
                                           op=dist.ReduceOp.AVG,
                                           async_op=True)
                    self._pending.append(work)

            self._hooks.append(p.register_post_accumulate_grad_hook(hook))

    def finish_grad_sync(self):
        """Wait for overlapped all-reduces (or do one synchronous pass if
        hooks are not installed). The exposed wait time is the NON-hidden
        part of gradient communication — the number that explains the
        1/2/4/8-GPU scaling curve (collective-time breakdown
        through the monitor; enable with MFX_MONITOR=debug|sidecar)."""
        import os
        import time

        from ..monitor import get_system_monitor

        # work.wait() on NCCL/RCCL is stream-ordered (host barely
        # blocks), so wall-clocking it reads ~0 even when comm
        # dominates. MFX_COMM_TIMING=1 brackets the wait with device
        # synchronizes: the measured delta is the comm still running
        # AFTER compute drained — the exposed (non-overlapped) tail.
        # Costs a sync per step, so it is opt-in for diagnosis runs.
        comm_timing = os.environ.get("MFX_COMM_TIMING") == "1" and \
            torch.cuda.is_available() and self.flat_grad.is_cuda
        if comm_timing:
            torch.cuda.synchronize()
        t0 = time.time()
        if self.zero and self.zero_world > 1:
            if self.sync_total == 0:    # pure-expert module: no comm
                self.last_comm_wait_ms = 0.0
                return
            s_, e_ = self.zero_shard
            if dist.get_backend(self.group or dist.group.WORLD) == "nccl":
                # each rank receives its averaged shard; (1-1/w) of the
                # grad volume moves vs 2(1-1/w) for all-reduce. Only
                # the sync region — expert grads stay owner-local.
                dist.reduce_scatter_tensor(
                    self.flat_grad[s_:e_],
                    self.flat_grad[:self.sync_total],
                    op=dist.ReduceOp.AVG, group=self.group)
            else:
                # gloo has no reduce_scatter_tensor: CPU-test emulation
                dist.all_reduce(self.flat_grad[:self.sync_total],
                                group=self.group, op=dist.ReduceOp.AVG)
            if comm_timing:
                torch.cuda.synchronize()
            get_system_monitor().gauge(
                "mfx.ddp.reduce_scatter_ms", (time.time() - t0) * 1000)
            self.last_comm_wait_ms = (time.time() - t0) * 1000
            return
        if self._pending:
            for work in self._pending:
                work.wait()
            self._pending.clear()
            for bi in self._bucket_done_count:
                self._bucket_done_count[bi] = 0
            if comm_timing:
                torch.cuda.synchronize()
            get_system_monitor().gauge(
                "mfx.ddp.allreduce_wait_ms", (time.time() - t0) * 1000)
        elif (dist.is_available() and dist.is_initialized()
              and dist.get_world_size(self.group) > 1):
            dist.all_reduce(self.flat_grad[:self.sync_end],
                            group=self.group, op=dist.ReduceOp.AVG)
            get_system_monitor().gauge(
                "mfx.ddp.allreduce_sync_ms", (time.time() - t0) * 1000)
        self.last_comm_wait_ms = (time.time() - t0) * 1000


class FusedAdamW(object):
    """AdamW over a FlatParamModel: one kernel call per step."""

    def __init__(self, flat_model, lr=3e-4, betas=(0.9, 0.95), eps=1e-8,
                 weight_decay=0.1, master_weights=False):
        fp = flat_model.flat_param
        self.flat = flat_model
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        s_, e_ = flat_model.zero_shard
        # ZeRO-1: fp32 state for this rank's shard (1/world of the
        # sync region) PLUS, under expert parallelism, the full
        # owner-local nosync segment
        n1 = e_ - s_
        seg = flat_model.local_seg if hasattr(flat_model, "local_seg") \
            else None
        n2 = (seg[1] - seg[0]) if seg else 0
        self._n1 = n1
        self.m = torch.zeros(n1 + n2, dtype=torch.float32,
                             device=fp.device)
        self.v = torch.zeros_like(self.m)
        if master_weights:
            self.master = torch.empty(n1 + n2, dtype=torch.float32,
                                      device=fp.device)
            self.master[:n1] = fp[s_:e_].float()
            if seg:
                self.master[n1:] = fp[seg[0]:seg[1]].float()
        else:
            self.master = None

    def step(self, grad_scale=1.0):
        self.step_count += 1
        s_, e_ = self.flat.zero_shard
        n1 = self._n1
        if n1:
            K.adamw_step(self.flat.flat_param[s_:e_],
                         self.flat.flat_grad[s_:e_], self.m[:n1],
                         self.v[:n1], self.step_count, self.lr,
                         self.beta1, self.beta2, self.eps,
                         self.weight_decay,
                         master=self.master[:n1]
                         if self.master is not None else None,
                         grad_scale=grad_scale)
        seg = getattr(self.flat, "local_seg", None)
        if seg:
            # owner-local expert segment: full state, local grads
            a, b = seg
            K.adamw_step(self.flat.flat_param[a:b],
                         self.flat.flat_grad[a:b], self.m[n1:],
                         self.v[n1:], self.step_count, self.lr,
                         self.beta1, self.beta2, self.eps,
                         self.weight_decay,
                         master=self.master[n1:]
                         if self.master is not None else None,
                         grad_scale=grad_scale)
        if self.flat.zero_world > 1 and self.flat.sync_total > 0:
            # publish the updated bf16 SYNC shard to every rank
            # (expert params are per-rank; nothing to gather there)
            fp = self.flat.flat_param
            st = self.flat.sync_total
            group = self.flat.group
            if dist.get_backend(group or dist.group.WORLD) == "nccl":
                dist.all_gather_into_tensor(fp[:st],
                                            fp[s_:e_].contiguous(),
                                            group=group)
            else:
                world = self.flat.zero_world
                ss = st // world
                shards = [torch.empty(ss, dtype=fp.dtype,
                                      device=fp.device)
                          for _ in range(world)]
                dist.all_gather(shards, fp[s_:e_].contiguous(),
                                group=group)
                for r, sh in enumerate(shards):
                    fp[r * ss:(r + 1) * ss].copy_(sh)

    def state_dict_tensors(self):
        """Shard tensors for @checkpoint (per-rank). Under ZeRO-1, m/v
        (and master) are this rank's 1/world shard — restore requires
        the same world size and rank mapping (the gang scheduler's
        stable rank->GPU pinning guarantees it on resume)."""
        out = {"flat_param": self.flat.flat_param, "adam_m": self.m,
               "adam_v": self.v}
        if self.master is not None:
            out["master"] = self.master
        return out


def init_process_group_from_env(backend=None):
    """env:// rendezvous (MASTER_ADDR/PORT, RANK, WORLD_SIZE) — set either
    by torchrun or by the gang scheduler (runtime._queue_gang)."""
    import os

    if dist.is_initialized():
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    if backend == "nccl":
        # bind the device BEFORE init: RCCL derives the communicator
        # device from the current device; late binds wedge rendezvous
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    from datetime import timedelta

    timeout_s = int(os.environ.get("MFX_RCCL_INIT_TIMEOUT", "300"))
    try:
        dist.init_process_group(backend=backend,
                                timeout=timedelta(seconds=timeout_s))
    except Exception as e:
        raise RuntimeError(
            "torch.distributed init failed (backend=%s, rank=%s/%s, "
            "addr=%s:%s): %s" % (
                backend, os.environ.get("RANK"),
                os.environ.get("WORLD_SIZE"),
                os.environ.get("MASTER_ADDR"),
                os.environ.get("MASTER_PORT"), e)) from e
