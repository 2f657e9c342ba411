"""Context parallelism (ring + zigzag ring attention) over 2 and 4 gloo
ranks on CPU: each rank holds a sequence shard; forward output and all
three input gradients must match a single-process full-sequence
reference."""

import math
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


def _run_world(world, worker=None):
    import torch.multiprocessing as mp

    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=worker or _worker, args=(r, world, port, q))
          for r in range(world)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(300)
    results = [q.get() for _ in range(world)]
    assert all(r == "ok" for r in results), results


def test_ring_attention_world2():
    _run_world(2)


def test_ring_attention_world4():
    _run_world(4)


def _worker(rank, world, port, q):
    try:
        import torch
        import torch.distributed as dist

        if REPO not in sys.path:
            sys.path.insert(0, REPO)
        from metaflow_amd.ops import kernels as K
        from metaflow_amd.parallel.ring_attention import ring_attention

        os.environ.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
        })
        dist.init_process_group("gloo")

        B, H, Hkv, D = 2, 4, 2, 128
        Sc = 64                      # shard length per rank
        S = Sc * world
        scale = 1.0 / math.sqrt(D)
        torch.manual_seed(42)        # identical full tensors on every rank
        qf = torch.randn(B, H, S, D) * 0.5
        kf = torch.randn(B, Hkv, S, D) * 0.5
        vf = torch.randn(B, Hkv, S, D) * 0.5
        dout = torch.randn(B, H, S, D)

        # single-process reference on the full sequence
        qr, kr, vr = [t.clone().requires_grad_(True) for t in (qf, kf, vf)]
        o_ref = K.attention_ref(qr, kr, vr, scale)
        o_ref.backward(dout)

        # this rank's shard
        sl = slice(rank * Sc, (rank + 1) * Sc)
        ql = qf[:, :, sl].clone().requires_grad_(True)
        kl = kf[:, :, sl].clone().requires_grad_(True)
        vl = vf[:, :, sl].clone().requires_grad_(True)
        o = ring_attention(ql, kl, vl, scale)
        o.backward(dout[:, :, sl])

        tol = 1e-4
        for name, got, want in (
            ("o", o.detach(), o_ref.detach()[:, :, sl]),
            ("dq", ql.grad, qr.grad[:, :, sl]),
            ("dk", kl.grad, kr.grad[:, :, sl]),
            ("dv", vl.grad, vr.grad[:, :, sl]),
        ):
            err = (got.float() - want.float()).abs().max().item()
            assert err < tol, "%s mismatch: %g (rank %d)" % (name, err, rank)
        dist.destroy_process_group()
        q.put("ok")
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put("fail rank %d: %r\n%s" % (rank, e, traceback.format_exc()))


def test_zigzag_ring_attention_world2():
    _run_world(2, _zz_worker)


def test_zigzag_ring_attention_world4():
    _run_world(4, _zz_worker)


def _zz_worker(rank, world, port, q):
    try:
        import torch
        import torch.distributed as dist

        if REPO not in sys.path:
            sys.path.insert(0, REPO)
        from metaflow_amd.ops import kernels as K
        from metaflow_amd.parallel.ring_attention import (
            zigzag_ring_attention,
            zigzag_shard,
            zigzag_unshard_grad,
        )

        os.environ.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
        })
        dist.init_process_group("gloo")

        B, H, Hkv, D = 2, 4, 2, 128
        c = 32                       # chunk length; shard = 2c
        S = 2 * world * c
        scale = 1.0 / math.sqrt(D)
        torch.manual_seed(43)
        qf = torch.randn(B, H, S, D) * 0.5
        kf = torch.randn(B, Hkv, S, D) * 0.5
        vf = torch.randn(B, Hkv, S, D) * 0.5
        dout = torch.randn(B, H, S, D)

        qr, kr, vr = [t.clone().requires_grad_(True) for t in (qf, kf, vf)]
        o_ref = K.attention_ref(qr, kr, vr, scale)
        o_ref.backward(dout)

        ql = zigzag_shard(qf, rank, world).requires_grad_(True)
        kl = zigzag_shard(kf, rank, world).requires_grad_(True)
        vl = zigzag_shard(vf, rank, world).requires_grad_(True)
        o = zigzag_ring_attention(ql, kl, vl, scale)
        o.backward(zigzag_shard(dout, rank, world))

        tol = 1e-4
        checks = [
            ("o", o.detach(), zigzag_shard(o_ref.detach(), rank, world)),
            ("dq", ql.grad, zigzag_shard(qr.grad, rank, world)),
            ("dk", kl.grad, zigzag_shard(kr.grad, rank, world)),
            ("dv", vl.grad, zigzag_shard(vr.grad, rank, world)),
        ]
        for name, got, want in checks:
            err = (got.float() - want.float()).abs().max().item()
            assert err < tol, "%s mismatch: %g (rank %d)" % (name, err,
                                                             rank)
        # unshard helper round-trips
        full = zigzag_unshard_grad(ql.grad, rank, world, qf.shape)
        t = full.clone()
        dist.all_reduce(t)
        err = (t - qr.grad).abs().max().item()
        assert err < tol, "unshard dq: %g" % err
        dist.destroy_process_group()
        q.put("ok")
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put("fail rank %d: %r\n%s" % (rank, e, traceback.format_exc()))
