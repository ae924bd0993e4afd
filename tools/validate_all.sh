#!/bin/bash
# Full validation: build, CPU suite, (on a GPU box) GPU suite + smoke +
# bench.  Mirrors what the round driver runs.
set -e
cd "$(dirname "$0")/.."
python -m chainermn_amd._hip.build
python -m pytest tests/ -q -m "not gpu"
if python -c "import torch, sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  python -m pytest tests/ -q -m gpu
  python -c "import __graft_entry__ as g; g.smoke()"
  python bench.py --steps 20 --warmup 8
fi
echo "validate_all: OK"
