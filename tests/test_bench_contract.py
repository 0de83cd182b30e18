"""Pin the bench.py JSON contract (the driver parses exactly this shape) on
the committed record and on the source, so a refactor cannot silently drop a
field the round-end harness or the judge reads."""
import inspect
import json
import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_TOP = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
    "roofline", "cpu_baseline",
]
REQUIRED_CONFIG = ["workload", "rows_per_gpu", "selectivity", "nvlink_domain_size",
                   "phases_ms", "alt_mode"]
REQUIRED_ROOFLINE = ["bound", "achieved", "peak", "unit", "frac", "traffic"]


def test_committed_record_shape():
    with open(os.path.join(REPO, "profiles", "r02_bench_n1.json")) as fh:
        rec = json.loads(fh.read().strip().splitlines()[-1])
    for k in REQUIRED_TOP:
        assert k in rec, k
    for k in REQUIRED_CONFIG:
        assert k in rec["config"], k
    for k in REQUIRED_ROOFLINE:
        assert k in rec["roofline"], k
    assert rec["metric"] == "joined rows/sec whole-node"
    assert rec["unit"] == "input rows/s"
    assert rec["scaling"] == "weak"
    assert rec["higher_is_better"] is True
    assert rec["n_gpus"] == 1 and rec["vs_baseline"] is None  # N=8 only
    assert rec["data"] == "synthetic" and rec["dtype"] == "int64"
    # whole-job value consistency: value = 2 * rows * N / step time
    expect = 2.0 * rec["config"]["rows_per_gpu"] * rec["n_gpus"] / (rec["ms_per_step"] / 1e3)
    assert abs(expect - rec["value"]) / rec["value"] < 1e-6
    # cpu_baseline contract (kind/cores/sample)
    cb = rec["cpu_baseline"]
    assert cb["kind"] in ("port", "reference") and cb["cores"] >= 1 and cb["sample"]


def test_bench_source_emits_required_fields():
    src = open(os.path.join(REPO, "bench.py")).read()
    for k in REQUIRED_TOP + REQUIRED_ROOFLINE + ["alt_mode", "workload"]:
        assert f'"{k}"' in src, k
    # the timed region must be barrier+sync bracketed and MAX-reduced
    assert "barrier_sync()" in src and "ReduceOp.MAX" in src
