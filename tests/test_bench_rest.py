"""The flagship bench's REST path must work end-to-end on CPU: client ->
control plane (subprocess) -> agent @reasoner -> app.ai() -> engine.
This is the exact topology the driver runs on the GPU (BASELINE configs
2/4), shrunk to the tiny model."""
import json
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def test_rest_bench_cpu_end_to_end():
    r = subprocess.run(
        [sys.executable, str(ROOT / "bench.py"), "--model", "tiny",
         "--steps", "1", "--warmup", "1", "--calls", "4",
         "--prompt-len", "48", "--gen-len", "4"],
        capture_output=True, text=True, timeout=300,
        env={**os.environ, "PYTHONPATH": str(ROOT),
             "MASTER_PORT": "29613"}, cwd=ROOT)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")][-1]
    out = json.loads(line)
    assert out["metric"] == "reasoner_calls_per_sec"
    assert out["value"] > 0
    assert out["config"]["path"] == "rest"
    assert out["config"]["p50_call_ms"] > 0
    assert out["n_gpus"] == 1
