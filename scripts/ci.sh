#!/usr/bin/env bash
# CI entry (the reference ran a full-stack docker workflow,
# .github/workflows/integration-tests.yml; this is the self-contained
# equivalent — no external Ollama/Redis needed).
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== build HIP extension (gfx950 cross-compile; no GPU required)"
python -c "from gridllm_amd.ops.build import build; print(build(force=True))"

echo "== CPU suite (control plane, gateway wire shapes, engine numerics,"
echo "   multi-process e2e + chaos, TP parity on gloo)"
python -m pytest tests -q -m "not gpu"

echo "== feature-matrix soak (full stack: 11 API surfaces, invariants)"
python scripts/soak.py --requests 176 --concurrency 16

echo "== fault-injection soak (worker SIGKILL/pause under live traffic)"
python scripts/chaos.py --duration 25 --workers 2 --fault-every 4

if python -c "import torch, sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  echo "== GPU suite (kernel numerics vs fp32 reference, hipGraph engine)"
  python -m pytest tests -q -m gpu
  echo "== smoke"
  python -c "import __graft_entry__ as g; g.smoke()"
fi
echo "CI OK"
