#!/usr/bin/env bash
# Local CI gates (what the round driver checks, runnable by hand).
#   ./scripts/ci.sh        # CPU-only gates
#   ./scripts/ci.sh gpu    # adds the GPU suite + smoke + a short bench
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== build (gfx950 cross-compile)"
PYTORCH_ROCM_ARCH=gfx950 python -c "import __graft_entry__ as g; g.build()"

echo "== CPU test suite"
python -m pytest tests/ -x -q -m "not gpu"

echo "== demo (deterministic, in-process)"
python scripts/demo_client.py --local >/dev/null && echo "demo ok"

if [[ "${1:-}" == "gpu" ]]; then
  echo "== GPU test suite"
  python -m pytest tests/ -x -q -m gpu
  echo "== smoke"
  python -c "import __graft_entry__ as g; g.smoke()"
  echo "== short bench"
  python bench.py --gpus 1 --steps 5 --warmup 2 --entries 1000000 --batch 1024
fi
echo "ALL GATES GREEN"
