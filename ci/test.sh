#!/usr/bin/env bash
# CI entry (reference ci/test.sh role): build the gfx950 extension + C ABI,
# then run the CPU test tier. GPU tier (pytest -m gpu) runs on MI355X boxes.
set -euo pipefail
cd "$(dirname "$0")/.."

python -c "import __graft_entry__ as g; g.build()"
python -m pytest tests/ -x -q -m "not gpu"

if python -c "import torch, sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
    python -m pytest tests/ -q -m gpu
    python -c "import __graft_entry__ as g; g.smoke()"
fi
