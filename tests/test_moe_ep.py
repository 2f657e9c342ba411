"""Expert-parallel Mixtral over 2 gloo ranks: token all-to-all dispatch,
owner-local expert grads (excluded from DP all-reduce), loss sanity."""

import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_mixtral_ep_world2():
    import torch.multiprocessing as mp

    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_worker, args=(r, port, q)) for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(300)
    results = [q.get() for _ in range(2)]
    assert all(r[0] == "ok" for r in results), results
    # both ranks converged on the same (allreduced) shared params but kept
    # their own experts
    assert results[0][1] != results[1][1], "expert params identical?"


def _worker(rank, port, q):
    try:
        import torch
        import torch.distributed as dist

        if REPO not in sys.path:
            sys.path.insert(0, REPO)
        from metaflow_amd.models.mixtral import (
            MixtralConfig,
            MixtralForCausalLM,
        )
        from metaflow_amd.parallel.ddp import FlatParamModel, FusedAdamW

        os.environ.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": "2",
        })
        dist.init_process_group("gloo")
        torch.manual_seed(7)  # same shared-param init on both ranks
        cfg = MixtralConfig.tiny(vocab=512, seq=128)
        model = MixtralForCausalLM(cfg)
        # perturb experts per rank (they are rank-local)
        torch.manual_seed(100 + rank)
        for layer in model.layers:
            for p in layer.moe.local_experts.parameters():
                p.data.add_(0.01 * torch.randn_like(p.data))
        assert layer.moe.ep_world == 2
        flat = FlatParamModel(model, bucket_mb=1)
        flat.install_overlap_hooks()
        opt = FusedAdamW(flat, lr=1e-3)

        torch.manual_seed(200 + rank)
        tok = torch.randint(0, cfg.vocab_size, (1, 129))
        losses = []
        for _ in range(3):
            flat.zero_grad()
            loss = model(tok[:, :-1], tok[:, 1:].contiguous())
            loss.backward()
            flat.finish_grad_sync()
            opt.step()
            losses.append(float(loss.detach()))
        assert all(l == l for l in losses), losses
        # shared (sync region) params must MATCH across ranks after steps
        sync_sum = float(
            flat.flat_param[:flat.sync_end].float().sum())
        t = torch.tensor([sync_sum])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        assert abs(float(t) - sync_sum) < 1e-3, "shared params diverged"
        expert_sig = float(
            flat.flat_param[flat.sync_end:].float().abs().sum())
        dist.destroy_process_group()
        q.put(("ok", round(expert_sig, 6)))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("fail: %r\n%s" % (e, traceback.format_exc()), 0))
