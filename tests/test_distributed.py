"""Multi-process (gloo, world_size=2) CPU tests of the distributed plumbing
bench.py uses at N>1: the all_reduce/all_gather exchange of partial blocks.
The engine-side merge semantics are covered on GPU by
test_engine_gpu.test_sharded_two_engines_merge; here we verify the torch
collective calls and block-layout handling run correctly multi-process.
"""
import os

import numpy as np
import pytest

torch = pytest.importorskip("torch")
import torch.distributed as dist  # noqa: E402
import torch.multiprocessing as mp  # noqa: E402


def _worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)

        # keyless block: [naggs sums][naggs counts][rowcount] f64 — all_reduce
        naggs = 2
        block = torch.tensor([1.0 * (rank + 1), 2.0 * (rank + 1),
                              10.0, 20.0, 30.0], dtype=torch.float64)
        dist.all_reduce(block)
        assert torch.allclose(block, torch.tensor([3.0, 6.0, 20.0, 40.0, 60.0],
                                                  dtype=torch.float64))

        # grouped blocks: equal-size uint8 buffers — all_gather then concat
        payload = np.zeros(64, dtype=np.uint8)
        payload[0] = rank + 7
        local = torch.from_numpy(payload)
        gathered = [torch.zeros(64, dtype=torch.uint8) for _ in range(world)]
        dist.all_gather(gathered, local)
        cat = torch.cat(gathered).numpy()
        assert cat[0] == 7 and cat[64] == 8

        # max-over-ranks elapsed (the timing contract)
        e = torch.tensor([1.0 + rank], dtype=torch.float64)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        assert float(e) == 2.0

        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as ex:  # pragma: no cover
        q.put((rank, f"fail: {ex!r}"))


def test_gloo_partial_exchange():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29000 + os.getpid() % 1000
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", (rank, status)


def _a2a_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        # key-sharded exchange transport: each rank holds `world` equal-size
        # destination blocks; the all-to-all delivers column `rank`.  gloo
        # has no all_to_all, so the CPU test uses the all_gather + slice
        # emulation bench.py's nccl path replaces with all_to_all_single.
        bb = 32
        send = np.zeros((world, bb), dtype=np.uint8)
        for d in range(world):
            send[d, 0] = 100 + 10 * rank + d   # marker: src, dst
        gathered = [torch.zeros(world * bb, dtype=torch.uint8)
                    for _ in range(world)]
        dist.all_gather(gathered, torch.from_numpy(send.reshape(-1)))
        recv = np.concatenate(
            [gathered[src].numpy().reshape(world, bb)[rank]
             for src in range(world)])
        # rank r must hold blocks (src=0,dst=r), (src=1,dst=r), ...
        for src in range(world):
            assert recv[src * bb] == 100 + 10 * src + rank
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as ex:  # pragma: no cover
        q.put((rank, f"fail: {ex!r}"))


def test_gloo_key_sharded_a2a_transport():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29000 + (os.getpid() + 17) % 1000
    procs = [ctx.Process(target=_a2a_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", (rank, status)
