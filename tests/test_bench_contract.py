"""bench.py contract tests: single-process and torchrun world-2 (gloo on
CPU — the same code path the driver uses with RCCL on GPUs)."""
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                 "dtype", "data", "config"}


def _last_json(text: str) -> dict:
    for line in reversed(text.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{text[-2000:]}")


def test_bench_single_process():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "tiny", "--calls", "2", "--prompt-len", "16",
         "--gen-len", "4"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json(out.stdout)
    assert REQUIRED_KEYS <= set(j)
    assert j["n_gpus"] == 1 and j["value"] > 0
    assert j["config"]["model"] == "tiny"
    assert j["config"]["parallelism"] == "dp1"


def test_bench_torchrun_world2():
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29687", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "0", "--model", "tiny", "--calls", "2",
         "--prompt-len", "16", "--gen-len", "4"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json(out.stdout)
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "dp2"
    assert j["config"]["global_batch"] == 4
