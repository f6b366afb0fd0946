"""Driver-contract test: bench.py emits one valid JSON line with the
required fields, on CPU, within a bounded time."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "2", "--warmup", "1", "--pods-per-step", "4"],
        capture_output=True, text=True, timeout=240, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"] == "Allocate() p50 latency"
    assert d["unit"] == "us"
    assert d["higher_is_better"] is False
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["vs_baseline"] is None
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    for key in ("model", "gpus_per_pod", "discovered_gpus",
                "admission_p50_us", "rpc_floor_us", "pods_per_s_total"):
        assert key in cfg, key
    assert cfg["discovered_gpus"] == 8
