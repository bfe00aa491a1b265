#!/usr/bin/env bash
# Full test sweep (reference parity: scripts/test-all.sh driving go test +
# pytest + lint).  GPU tier runs only where an MI355X is visible.
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== build native (hipcc gfx950 cross-compile + C++ ext) =="
python agentfield_amd/build.py

echo "== C++ SDK =="
make -s -C sdk/cpp

echo "== sanitizer + soak tier (ASAN native build; threaded CP soak) =="
python -m pytest tests/test_sanitizer.py -q

echo "== CPU test tier (incl. gloo multi-process) =="
python -m pytest tests/ -q -m "not gpu" --deselect tests/test_sanitizer.py

if python -c "import torch; raise SystemExit(0 if torch.cuda.is_available() else 1)" 2>/dev/null; then
  echo "== GPU test tier (MI355X) =="
  python -m pytest tests/ -q -m gpu
else
  echo "== GPU tier skipped (no GPU visible) =="
fi
