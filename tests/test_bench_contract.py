"""CPU-side guards on bench.py's contract pieces: the cpu_baseline leg (runs
the oracle CLI) and the committed roofline-traffic file."""
import importlib.util
import json
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _bench():
    spec = importlib.util.spec_from_file_location(
        "bench", os.path.join(REPO, "bench.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def test_cpu_baseline_object():
    b = _bench()
    cb = b.cpu_baseline("q1", 2_000_000)
    assert cb["kind"] == "port" and cb["cores"] == 1
    assert cb["unit"] == "rows/s" and cb["value"] > 0
    assert "scalar" in cb["sample"]


def test_cpu_baseline_q3():
    b = _bench()
    cb = b.cpu_baseline("q3", 1_000_000)
    assert cb["value"] > 0


def test_traffic_file_parses():
    b = _bench()
    t = b.load_traffic("tpch_q1", 100)
    assert t is not None and 20e9 < t < 30e9  # ≈ algorithmic 22.8 GB
    assert b.load_traffic("tpch_q1", 10) is None  # sf mismatch → null
    assert b.load_traffic("nope", 100) is None


def test_committed_bench_snapshots_schema():
    for f in ("r01_bench_q1.json", "r01_bench_q3.json"):
        with open(os.path.join(REPO, "profiles", f)) as fh:
            d = json.load(fh)
        for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling", "dtype",
                  "data", "config", "roofline"):
            assert k in d, k
        assert d["roofline"]["bound"] == "hbm"
        assert 0 < d["roofline"]["frac"] < 1


def test_cpu_baseline_sharded():
    """shards>1 = one CLI process per shard (DataNode deployment analog):
    whole-job rows/s over max-shard time, cores = shards, and roughly
    additive throughput vs the 1-core run on an idle host."""
    b = _bench()
    one = b.cpu_baseline("q1", 2_000_000)
    four = b.cpu_baseline("q1", 2_000_000, shards=4)
    assert four["cores"] == 4 and four["kind"] == "port"
    assert "4 shard processes" in four["sample"]
    # parallel shards must beat one core (loose: >=1.5x, CI boxes share)
    assert four["value"] > 1.5 * one["value"]


def test_cpu_baseline_shard_rounding():
    """sample_rows not divisible by shards is rounded down (generator
    requires rows % nranks == 0)."""
    b = _bench()
    cb = b.cpu_baseline("q1", 1_000_001, shards=4)
    assert cb["value"] > 0 and "1000000 " in cb["sample"]
