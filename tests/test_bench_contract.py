"""bench.py contract test: the driver depends on the one-line JSON output."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--config", "kind-fake"],
        cwd=REPO, capture_output=True, text=True, timeout=280,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"].startswith("fractional pods/GPU + p50 Allocate()")
    assert d["n_gpus"] == 1
    assert d["steps"] == 3 and d["warmup"] == 1
    assert d["value"] > 0 and d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["p50_allocate_us"] > 0
    assert d["ms_per_step"] > 0
    assert "config" in d and d["config"]["pods_per_gpu"] > 0


@pytest.mark.timeout(300)
def test_bench_scenarios_run():
    for scenario in ("whole-gpu", "mem-fraction", "compute-fraction"):
        out = subprocess.run(
            [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
             "--config", scenario],
            cwd=REPO, capture_output=True, text=True, timeout=280,
        )
        assert out.returncode == 0, (scenario, out.stderr[-2000:])
        d = json.loads(out.stdout.strip().splitlines()[-1])
        assert d["config"]["scenario"] == scenario
