"""Tensor parallelism over 2 gloo ranks on CPU: a TP-sharded Llama built
from an unsharded reference model must produce the same loss and the
same gradient for every weight shard."""

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


def _run(sp):
    import torch.multiprocessing as mp

    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_worker, args=(r, port, q, sp))
          for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(300)
    results = [q.get() for _ in range(2)]
    assert all(r == "ok" for r in results), results


def test_tp_llama_world2():
    _run(sp=False)


def test_tp_llama_sequence_parallel_world2():
    """TP + Megatron sequence parallelism: norm/residual activations
    sequence-sharded; same exactness bar as plain TP (incl. the
    sp_sync_grads pass for the SP-partial params)."""
    _run(sp=True)


def _worker(rank, port, q, sp=False):
    try:
        import torch
        import torch.distributed as dist

        if REPO not in sys.path:
            sys.path.insert(0, REPO)
        from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
        from metaflow_amd.models.llama_tp import TPLlamaForCausalLM
        from metaflow_amd.parallel.tp import (
            shard_cols,
            shard_gate_up_rows,
            shard_qkv_rows,
            shard_rows,
        )

        os.environ.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": "2",
        })
        dist.init_process_group("gloo")
        world = 2

        torch.manual_seed(21)
        cfg = LlamaConfig.tiny(vocab=256, seq=64)
        cfg.num_heads, cfg.num_kv_heads = 4, 2
        ref = LlamaForCausalLM(cfg)
        tp = TPLlamaForCausalLM.from_full_model(ref, dist.group.WORLD,
                                                sequence_parallel=sp)

        torch.manual_seed(77)
        tok = torch.randint(0, cfg.vocab_size, (2, 65))
        inp, tgt = tok[:, :-1], tok[:, 1:].contiguous()

        loss_ref = ref(inp, tgt)
        loss_ref.backward()
        loss = tp(inp, tgt)
        loss.backward()
        if sp:
            from metaflow_amd.parallel.tp import sp_sync_grads

            sp_sync_grads(tp, dist.group.WORLD)

        assert abs(float(loss) - float(loss_ref)) < 2e-3, \
            (float(loss), float(loss_ref))

        def check(name, got, want):
            denom = want.float().abs().max().item() + 1e-6
            err = (got.float() - want.float()).abs().max().item() / denom
            assert err < 6e-2, "%s grad mismatch %g (rank %d)" % (
                name, err, rank)

        nq, nkv, hd = cfg.num_heads, cfg.num_kv_heads, cfg.head_dim
        vshard = cfg.vocab_size // world
        check("embed", tp.embed.weight.grad, ref.embed.weight.grad)
        check("final_norm", tp.final_norm.weight.grad,
              ref.final_norm.weight.grad)
        check("lm_head", tp.lm_head.weight.grad,
              ref.lm_head.weight.grad[rank * vshard:(rank + 1) * vshard])
        for li, (tl, fl) in enumerate(zip(tp.layers, ref.layers)):
            check("l%d.qkv" % li, tl.qkv_proj.weight.grad,
                  shard_qkv_rows(fl.qkv_proj.weight.grad, rank, world,
                                 nq, nkv, hd))
            check("l%d.o" % li, tl.o_proj.weight.grad,
                  shard_cols(fl.o_proj.weight.grad, rank, world))
            check("l%d.gu" % li, tl.gate_up_proj.weight.grad,
                  shard_gate_up_rows(fl.gate_up_proj.weight.grad, rank,
                                     world))
            check("l%d.down" % li, tl.down_proj.weight.grad,
                  shard_cols(fl.down_proj.weight.grad, rank, world))
            check("l%d.in_norm" % li, tl.input_norm.weight.grad,
                  fl.input_norm.weight.grad)
            check("l%d.post_norm" % li, tl.post_norm.weight.grad,
                  fl.post_norm.weight.grad)
        dist.destroy_process_group()
        q.put("ok")
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put("fail rank %d: %r\n%s" % (rank, e, traceback.format_exc()))


def test_tp_flow_through_scheduler(tmp_datastore):
    """@torch_parallel(tensor_parallel=2) gang through the real
    scheduler: TP model trains, loss falls, tp ranks agree."""
    from .test_runtime import latest_run_id, read_artifact, run_flow

    run_flow("tp_flow.py", tmp_datastore, "run", timeout=420)
    run_id = latest_run_id(tmp_datastore, "TPFlow")
    losses = read_artifact(tmp_datastore, "TPFlow", run_id, "join",
                           "losses")
    assert len(losses) == 2 and losses[0] == losses[1]


def test_tp2_dp2_world4():
    """Composed grid at world 4: tp=2 inside dp=2. Each dp replica sees a
    different batch; after the dp-group flat all-reduce every rank's
    shard grads must equal the batch-averaged reference grads."""
    import torch.multiprocessing as mp

    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_grid_worker, args=(r, port, q))
          for r in range(4)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(420)
    results = [q.get() for _ in range(4)]
    assert all(r == "ok" for r in results), results


def _grid_worker(rank, port, q):
    try:
        import torch
        import torch.distributed as dist

        if REPO not in sys.path:
            sys.path.insert(0, REPO)
        from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
        from metaflow_amd.parallel.ddp import FlatParamModel
        from metaflow_amd.models.llama_tp import TPLlamaForCausalLM

        os.environ.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": "4",
        })
        dist.init_process_group("gloo")
        tp_deg = 2
        tp_group = dp_group = None
        for start in (0, 2):  # consecutive ranks share a tp group
            g = dist.new_group([start, start + 1])
            if start <= rank < start + 2:
                tp_group = g
        for idx in (0, 1):    # same shard index across replicas = dp
            g = dist.new_group([idx, idx + 2])
            if rank % tp_deg == idx:
                dp_group = g

        torch.manual_seed(21)
        cfg = LlamaConfig.tiny(vocab=256, seq=64)
        cfg.num_heads, cfg.num_kv_heads = 4, 2
        ref = LlamaForCausalLM(cfg)
        tp = TPLlamaForCausalLM.from_full_model(ref, tp_group)
        flat = FlatParamModel(tp, bucket_mb=1, group=dp_group)

        dp_rank = rank // tp_deg
        torch.manual_seed(101 + dp_rank)   # one batch per dp replica
        tok = torch.randint(0, cfg.vocab_size, (2, 65))
        inp, tgt = tok[:, :-1], tok[:, 1:].contiguous()

        loss = tp(inp, tgt)
        loss.backward()
        flat.finish_grad_sync()            # dp-average over dp_group

        # reference: average the two per-batch gradient sets
        grads = {}
        for b in (0, 1):
            refb = LlamaForCausalLM(cfg)
            refb.load_state_dict(ref.state_dict())
            torch.manual_seed(101 + b)
            tb = torch.randint(0, cfg.vocab_size, (2, 65))
            lb = refb(tb[:, :-1], tb[:, 1:].contiguous())
            lb.backward()
            for n, p in refb.named_parameters():
                grads.setdefault(n, []).append(p.grad.float())
        avg = {n: (g[0] + g[1]) / 2 for n, g in grads.items()}

        from metaflow_amd.parallel.tp import shard_qkv_rows

        tpr = dist.get_rank(tp_group)
        l0 = avg["layers.0.qkv_proj.weight"]
        want = shard_qkv_rows(l0, tpr, tp_deg, cfg.num_heads,
                              cfg.num_kv_heads, cfg.head_dim)
        got = tp.layers[0].qkv_proj.weight.grad.float()
        denom = want.abs().max().item() + 1e-6
        err = (got - want).abs().max().item() / denom
        assert err < 8e-2, "qkv dp-averaged grad mismatch %g" % err
        # replicated param too
        wn = avg["final_norm.weight"]
        gn = tp.final_norm.weight.grad.float()
        errn = (gn - wn).abs().max().item() / (wn.abs().max() + 1e-6)
        assert errn < 8e-2, "norm dp-averaged grad mismatch %g" % errn
        dist.destroy_process_group()
        q.put("ok")
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put("fail rank %d: %r\n%s" % (rank, e, traceback.format_exc()))
