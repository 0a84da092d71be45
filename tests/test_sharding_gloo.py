"""Multi-process sharding test (CPU, gloo, world_size=2).

Mirrors the multi-GPU execution model of bench.py: blocks shard across ranks
(SURVEY.md §8e — blocks are independent; the reference feeds them to
independent workers, storage_search.go:1035-1067), each rank scans its shard,
and the only exchange is an all_reduce(SUM) of hit counts.  On GPU the same
code path runs over RCCL; here gloo validates the sharding and the reduce."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

FILTER = '{"type":"phrase","field":"_msg","phrase":"the stream 1 and"}'


def _worker(rank, world, part_dir, port, out):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import sys
        sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        from victorialogs_amd import OracleScanner

        orc = OracleScanner(part_dir)
        nblocks = orc.blocks
        lo = rank * nblocks // world
        hi = (rank + 1) * nblocks // world
        hits, _ = orc.scan(FILTER, lo=lo, hi=hi)
        orc.close()

        t = torch.tensor([hits], dtype=torch.int64)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        out[rank] = int(t.item())
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_block_sharding_allreduce(gen_part):
    from victorialogs_amd import OracleScanner

    orc = OracleScanner(gen_part)
    expected, _ = orc.scan(FILTER)
    orc.close()
    assert expected > 0

    ctx = mp.get_context("spawn")
    out = ctx.Manager().dict()
    port = 29781
    procs = [ctx.Process(target=_worker, args=(r, 2, gen_part, port, out))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=100)
        assert p.exitcode == 0
    assert out[0] == expected and out[1] == expected
