"""BASELINE config 3 (3-agent DAG + DID/VC + prefix cache) must work
end-to-end; the GPU bench (tools/dag_bench.py) runs this exact topology
on the MI355X with llama-3-8b."""
import json
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def test_dag_bench_cpu_end_to_end():
    r = subprocess.run(
        [sys.executable, str(ROOT / "tools" / "dag_bench.py"),
         "--model", "tiny", "--calls", "4", "--gen", "6",
         "--cp-port", "18240"],
        capture_output=True, text=True, timeout=300,
        env={**os.environ, "PYTHONPATH": str(ROOT)}, cwd=ROOT)
    assert r.returncode == 0, r.stderr[-2000:]
    out = json.loads([ln for ln in r.stdout.splitlines()
                      if ln.startswith("{")][-1])
    assert out["dag_nodes"] >= 3
    assert out["vc_chain_len"] >= 3 and out["vc_chain_valid"]
    assert out["prefix_cache"]["cache_hits"] > 0
    assert out["prefix_cache"]["scheduler"] == "NativeSchedulerAdapter"
