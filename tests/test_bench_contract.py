"""Guard the driver contracts: bench.py's JSON line and __graft_entry__.

The round driver launches `python bench.py --gpus N --steps K --warmup W`
and parses EXACTLY ONE JSON line from rank 0 with a fixed key set; it also
calls __graft_entry__.build() each round. A contract break here silently
invalidates the round's measurements — pin it.
"""

import json
import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=280,
    )
    assert r.returncode == 0, r.stderr[-800:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout  # exactly ONE JSON line
    d = json.loads(lines[0])
    required = {
        "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
        "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
    }
    assert required <= set(d), required - set(d)
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert {"model", "global_batch", "seq_len", "parallelism"} <= set(d["config"])
    assert d["value"] > 0 and d["ms_per_step"] > 0


def test_graft_entry_surface():
    import __graft_entry__

    assert callable(__graft_entry__.build)
    assert callable(__graft_entry__.smoke)
