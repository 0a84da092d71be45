"""Multi-process sharding through the PRODUCT library.

tests/test_sharding_gloo.py validates the sharding/all-reduce shape with
the oracle; these tests drive the REAL product code paths on CPU:

- world-2 gloo ranks each staging + scanning their block shard through the
  emulated product pipeline (real vql_api.cpp staging, real per-row device
  code; only the wavefront shells are serial loops), all_reduce of hits;
- bench.py itself under torch.distributed.run with 2 ranks (its actual
  distributed branches: init_process_group, barriers, MAX/SUM all_reduces,
  selectivity check aggregation) against the emu build.

On a GPU box the same branches run over RCCL (profiles/r02 records a
world-2 NCCL run on one device)."""

import json
import os
import socket
import subprocess
import sys

import pytest
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EMU = os.path.join(ROOT, "tools", "host_emu", "libvlogsql_emu.so")

pytestmark = pytest.mark.skipif(not os.path.exists(EMU),
                                reason="emu lib not built (make emu)")

FILTER = '{"type":"phrase","field":"_msg","phrase":"the stream 1 and"}'


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(rank, world, part_dir, port, out):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["VQL_LIB"] = EMU  # before the product lib is loaded
    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        sys.path.insert(0, ROOT)
        from victorialogs_amd import Filter, Part, Stage

        part = Part(part_dir)
        nblocks = part.blocks
        lo = rank * nblocks // world
        hi = (rank + 1) * nblocks // world
        filt = Filter(FILTER)
        st = Stage(part, filt, lo=lo, hi=hi)
        hits = st.scan()
        # per-block popcounts of the shard must sum to the shard's hits
        # (`| stats count()` fast path, block_result.go:403-413)
        block_hits = st.fetch_block_hits(hi - lo)
        assert sum(block_hits) == hits
        st.close()
        filt.close()
        part.close()

        t = torch.tensor([hits], dtype=torch.int64)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        out[rank] = int(t.item())
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_product_block_sharding_allreduce(gen_part):
    from victorialogs_amd import OracleScanner

    orc = OracleScanner(gen_part)
    expected, _ = orc.scan(FILTER)
    orc.close()
    assert expected > 0

    ctx = mp.get_context("spawn")
    out = ctx.Manager().dict()
    port = free_port()
    procs = [ctx.Process(target=_worker, args=(r, 2, gen_part, port, out))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=280)
        assert p.exitcode == 0
    assert out[0] == expected and out[1] == expected


@pytest.mark.timeout(600)
def test_bench_distributed_cpu(tmp_path):
    """bench.py's own world-2 distributed code path, end to end."""
    env = dict(os.environ, VQL_LIB=EMU, VQL_DATA_DIR=str(tmp_path),
               MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--rows", "200000", "--steps", "2", "--warmup", "1",
         "--skip-cpu-baseline"],
        env=env, cwd=ROOT, capture_output=True, text=True, timeout=550)
    assert r.returncode == 0, r.stdout[-4000:] + "\n" + r.stderr[-4000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["n_gpus"] == 2
    # whole-job aggregate: both ranks' 200k rows, matched == scanned for the
    # all-match headline; the full-size selectivity check aggregated exactly
    assert res["config"]["matched_rows_per_pass"] == 400000
    assert res["selectivity_check"]["ok"]
    assert res["selectivity_check"]["expected_matches"] > 0
    assert res["value"] > 0


def test_fetch_block_hits_vs_oracle(gen_part):
    """vql_fetch_block_hits per-block counts == oracle per-block counts
    (VERDICT r01 weak item 4), via the emulated product pipeline."""
    out_code = """
import sys
sys.path.insert(0, ".")
from victorialogs_amd import Filter, Part, Stage, OracleScanner
part_dir = %r
filter_json = %r
part = Part(part_dir)
filt = Filter(filter_json)
st = Stage(part, filt)
hits = st.scan()
bh = st.fetch_block_hits(part.blocks)
orc = OracleScanner(part_dir)
exp = [orc.scan(filter_json, lo=i, hi=i + 1)[0] for i in range(part.blocks)]
orc.close()
assert bh == exp, (bh[:10], exp[:10])
assert sum(bh) == hits and hits > 0
print("block hits OK", hits)
""" % (gen_part, FILTER)
    env = dict(os.environ, VQL_LIB=EMU)
    r = subprocess.run([sys.executable, "-c", out_code], env=env, cwd=ROOT,
                       capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout[-2000:] + "\n" + r.stderr[-2000:]
    assert "block hits OK" in r.stdout
