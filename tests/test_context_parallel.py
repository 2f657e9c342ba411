"""End-to-end context-parallel Llama on CPU (gloo, world 2): two ranks
each hold half the sequence; the cp-averaged loss and cp-averaged
parameter gradients must match a single-process full-sequence model."""

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


def test_llama_context_parallel_world2():
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
    assert all(r == "ok" for r in results), results


def _worker(rank, port, q):
    try:
        import torch
        import torch.distributed as dist

        if REPO not in sys.path:
            sys.path.insert(0, REPO)
        from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM

        os.environ.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": "2",
        })
        dist.init_process_group("gloo")

        S, Sc = 128, 64
        torch.manual_seed(11)      # identical params on both ranks
        cfg = LlamaConfig.tiny(vocab=256, seq=S)
        ref = LlamaForCausalLM(cfg)
        torch.manual_seed(11)
        model = LlamaForCausalLM(cfg, cp_group=dist.group.WORLD)

        torch.manual_seed(77)      # same batch everywhere
        tok = torch.randint(0, cfg.vocab_size, (1, S + 1))
        inp, tgt = tok[:, :-1], tok[:, 1:].contiguous()

        # single-process full-sequence reference
        loss_ref = ref(inp, tgt)
        loss_ref.backward()

        # this rank's shard
        sl = slice(rank * Sc, (rank + 1) * Sc)
        loss = model(inp[:, sl], tgt[:, sl].contiguous())
        loss.backward()

        # cp-average of local losses == global mean loss
        t = loss.detach().clone()
        dist.all_reduce(t)
        t /= 2
        assert abs(float(t) - float(loss_ref)) < 1e-3, \
            "loss %g vs ref %g" % (float(t), float(loss_ref))

        # cp-average of grads == full-sequence grads (the DDP all-reduce
        # contract from LlamaForCausalLM's docstring)
        worst = ("", 0.0)
        for (name, p), (_, pr) in zip(model.named_parameters(),
                                      ref.named_parameters()):
            if p.grad is None:
                assert pr.grad is None or pr.grad.abs().max() == 0, name
                continue
            g = p.grad.float().clone()
            dist.all_reduce(g)
            g /= 2
            denom = pr.grad.float().abs().max().item() + 1e-6
            err = (g - pr.grad.float()).abs().max().item() / denom
            if err > worst[1]:
                worst = (name, err)
        assert worst[1] < 5e-2, "grad mismatch %s: %g" % worst
        dist.destroy_process_group()
        q.put("ok")
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put("fail rank %d: %r\n%s" % (rank, e, traceback.format_exc()))
