"""The bench.py contract itself, exercised at a small scale on the GPU box:
one JSON line with the required fields, value consistent with ms_per_step,
parity check against the oracle enabled."""
import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_contract_small():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--scale", "16", "--steps", "3",
         "--warmup", "1", "--cpu-baseline", "0", "--check", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
                "csr_build_ms", "roofline"):
        assert key in d, key
    assert d["n_gpus"] == 1 and d["steps"] == 3 and d["warmup"] == 1
    assert d["unit"] == "edges/s"
    E = d["config"]["edges"]
    assert abs(d["value"] - E / (d["ms_per_step"] / 1e3)) / d["value"] < 1e-6
    assert d["roofline"]["bound"] == "hbm" and d["roofline"]["peak"] == 8000.0
    assert d["parity_linf_vs_oracle"] <= 1e-6
