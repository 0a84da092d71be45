"""End-to-end CPU run of the PRODUCT pipeline via the emulation build
(tools/host_emu/libvlogsql_emu.so): the real filter compile + real staging
(vql_api.cpp) + the real per-row device code (scan_rowops.h), with only the
HIP runtime and the wavefront kernel shells replaced by serial loops.

This catches staging/descriptor bugs on the CPU that otherwise only surface
on a GPU box; the real GPU parity suite still runs the same batteries with
the true kernels."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EMU = os.path.join(ROOT, "tools", "host_emu", "libvlogsql_emu.so")

pytestmark = pytest.mark.skipif(not os.path.exists(EMU),
                                reason="emu lib not built (make emu)")


def run_in_emu(code):
    """Runs python code in a subprocess with VQL_LIB pointing at the emu
    build (the product lib is cached per process, so isolation is needed)."""
    env = dict(os.environ, VQL_LIB=EMU)
    r = subprocess.run([sys.executable, "-c", code], env=env, cwd=ROOT,
                       capture_output=True, text=True, timeout=1200)
    assert r.returncode == 0, r.stdout[-4000:] + "\n" + r.stderr[-4000:]
    return r.stdout


def test_emu_batteries(tmp_path):
    """Both filter batteries, bit-identical to the oracle, through the
    emulated product pipeline."""
    out = run_in_emu("""
import json, sys
sys.path.insert(0, ".")
from tests.conftest import FILTERS, TYPED_FILTERS
from tests.test_gpu_parity import assert_parity
import tests.conftest as cf

# build the same session fixtures conftest would
import tempfile
from victorialogs_amd import generate_part

class TF:
    def mktemp(self, x):
        import pathlib
        return pathlib.Path(tempfile.mkdtemp())

gen = cf.gen_part.__wrapped__(TF())
typed = cf.typed_part.__wrapped__(TF())
n = 0
for f in FILTERS:
    assert_parity(gen, f)
    n += 1
for f in TYPED_FILTERS:
    assert_parity(typed, f)
    n += 1
print("emu battery OK", n)
""")
    assert "emu battery OK" in out


def test_emu_reference_fixtures():
    """All committed reference fixtures through the emulated pipeline."""
    out = run_in_emu("""
import sys
sys.path.insert(0, ".")
from tests.test_reference_filter_fixtures import (
    test_reference_fixtures_gpu, test_reference_ts_fixtures_gpu)
import tempfile, pathlib
test_reference_fixtures_gpu(pathlib.Path(tempfile.mkdtemp()))
test_reference_ts_fixtures_gpu(pathlib.Path(tempfile.mkdtemp()))
print("emu fixtures OK")
""")
    assert "emu fixtures OK" in out


def test_emu_fuzz_trees():
    """The differential fuzz trees through the emulated pipeline."""
    out = run_in_emu("""
import json, random, sys, tempfile, pathlib
sys.path.insert(0, ".")
import tests.conftest as cf
from tests.test_gpu_parity import assert_parity
from tests.test_gpu_fuzz import (GEN_PHRASES, GEN_FIELDS_STR, GEN_FIELDS_NUM,
                                 GEN_REGEXES, TYPED_PHRASES, TYPED_FIELDS_STR,
                                 TYPED_FIELDS_NUM, TYPED_REGEXES, random_tree)

class TF:
    def mktemp(self, x):
        return pathlib.Path(tempfile.mkdtemp())

gen = cf.gen_part.__wrapped__(TF())
typed = cf.typed_part.__wrapped__(TF())
for seed in range(4):
    rng = random.Random(1000 + seed)
    for _ in range(10):
        t = random_tree(rng, GEN_PHRASES, GEN_FIELDS_STR, GEN_FIELDS_NUM,
                        GEN_REGEXES, depth=3)
        assert_parity(gen, json.dumps(t))
    rng = random.Random(2000 + seed)
    for _ in range(10):
        t = random_tree(rng, TYPED_PHRASES, TYPED_FIELDS_STR,
                        TYPED_FIELDS_NUM, TYPED_REGEXES, depth=3)
        assert_parity(typed, json.dumps(t))
print("emu fuzz OK")
""")
    assert "emu fuzz OK" in out


def test_emu_gather_and_drivers(tmp_path):
    """Gather, multi-part staging, vql_scan_query and the bloom build
    through the emulated pipeline (their gpu-marked twins run the same code
    against the real kernels)."""
    out = run_in_emu("""
import sys, tempfile, pathlib
sys.path.insert(0, ".")
import tests.conftest as cf
import tests.test_gpu_gather as gg
import tests.test_gpu_parity as gp
import tests.test_gpu_bloom_build as gb

class TF:
    def mktemp(self, x):
        return pathlib.Path(tempfile.mkdtemp())

gen = cf.gen_part.__wrapped__(TF())
typed = cf.typed_part.__wrapped__(TF())

# gather suite over the typed part
from victorialogs_amd import Filter, Part, Stage
part = Part(typed)
filt = Filter('{"type":"phrase","field":"lvl","phrase":"error"}')
st = Stage(part, filt, device=0)
st.scan()

class _S:  # mimic the module fixture
    pass

gg.test_gather_rowids(st)
gg.test_gather_string_column(st)
gg.test_gather_const_column(st)
gg.test_gather_numeric_columns(st)
gg.test_gather_missing_column(st)
st.close(); filt.close(); part.close()
gg.test_gather_multiblock(typed)

gp.test_multipart_stage(gen, typed)
gp.test_scan_query_driver(gen, typed)
gp.test_multichunk_block(pathlib.Path(tempfile.mkdtemp()))
gp.test_long_rows_global_fallback(pathlib.Path(tempfile.mkdtemp()))

gb.test_bloom_build_simple()
gb.test_bloom_build_unicode()
gb.test_bloom_build_edge_cases()
print("emu gather/drivers OK")
""")
    assert "emu gather/drivers OK" in out


def test_emu_bench_contract(tmp_path):
    """bench.py end to end (small rows) under the emulated pipeline: the
    printed JSON line must satisfy the driver contract."""
    env = dict(os.environ, VQL_LIB=EMU,
               VQL_DATA_DIR=str(tmp_path / "bench_data"))
    r = subprocess.run(
        [sys.executable, "bench.py", "--rows", "200000", "--steps", "2",
         "--warmup", "1", "--skip-cpu-baseline"],
        env=env, cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    line = json.loads(r.stdout.strip().splitlines()[-1])
    for key in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config", "roofline", "cpu_baseline"]:
        assert key in line, key
    assert line["metric"] == "matched-rows/sec"
    assert line["n_gpus"] == 1 and line["scaling"] == "weak"
    assert line["config"]["workload"].startswith("100M rows") or \
        "phrase" in line["config"]["filter"]
    assert line["roofline"]["bound"] == "hbm"
    assert line["value"] > 0


def test_emu_random_parts():
    """Randomized PARTS (random typed columns incl. contaminated type mixes)
    x random filters through the emulated pipeline vs the oracle — covers
    encoder-priority and per-type staging corners beyond the fixed
    fixtures."""
    out = run_in_emu("""
import json, random, sys, tempfile
sys.path.insert(0, ".")
from victorialogs_amd import write_custom_part
from tests.test_gpu_parity import assert_parity
from tests.test_gpu_fuzz import random_tree

def rand_value(rng, kind):
    if kind == "u8": return str(rng.randrange(0, 256))
    if kind == "u16": return str(rng.randrange(0, 1 << 16))
    if kind == "u32": return str(rng.randrange(0, 1 << 32))
    if kind == "u64": return str(rng.randrange(0, 1 << 64))
    if kind == "i64": return str(rng.randrange(-1 << 62, 1 << 62))
    if kind == "f64": return repr(rng.uniform(-1e6, 1e6))
    if kind == "ip": return "%d.%d.%d.%d" % tuple(
        rng.randrange(0, 256) for _ in range(4))
    if kind == "iso": return "20%02d-%02d-%02dT%02d:%02d:%02d.%03dZ" % (
        rng.randrange(0, 60), rng.randrange(1, 13), rng.randrange(1, 29),
        rng.randrange(0, 24), rng.randrange(0, 60), rng.randrange(0, 60),
        rng.randrange(0, 1000))
    if kind == "dict": return rng.choice(["alpha", "beta", "gamma", "delta"])
    return "".join(rng.choice("ab 01_.-!") for _ in range(rng.randrange(0, 20)))

KINDS = ["u8", "u16", "u32", "u64", "i64", "f64", "ip", "iso", "dict", "word"]
PHRASES = ["alpha", "1", "12", "255", "ab", "0", "-", ".", "", "beta b"]
REGEXES = ["al.*a", "^1\\\\d", "a|b", "(be|ga)ta", "\\\\d+\\\\.\\\\d+",
           "1.2", "[0-9]{2,3}$"]
n = 0
for seed in range(300, 312):
    rng = random.Random(seed)
    rows = rng.randrange(1, 300)
    cols, names = [], []
    for c in range(rng.randrange(1, 5)):
        kind = rng.choice(KINDS)
        mix = rng.random() < 0.25
        vals = [rand_value(rng, rng.choice(KINDS))
                if (mix and rng.random() < 0.1) else rand_value(rng, kind)
                for _ in range(rows)]
        names.append("c%d_%s" % (c, kind))
        cols.append({"name": names[-1], "values": vals})
    ts0 = rng.randrange(-1 << 60, 1 << 60)
    d = tempfile.mkdtemp() + "/p"
    write_custom_part(d, {"blocks": [{
        "stream": 0,
        "timestamps": sorted(ts0 + rng.randrange(0, 1 << 40)
                             for _ in range(rows)),
        "columns": cols}]})
    for _ in range(8):
        t = random_tree(rng, PHRASES, names + ["missing"], names, REGEXES,
                        depth=3)
        assert_parity(d, json.dumps(t))
        n += 1
print("emu random parts OK", n)
""")
    assert "emu random parts OK" in out
