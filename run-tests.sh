#!/bin/bash
# CPU suite everywhere; GPU suite when an MI355X is visible
# (the reference's run-tests.sh counterpart).
set -e
cd "$(dirname "$0")"
python -m pytest tests/ -q -m "not gpu" "$@"
if python -c "import torch, sys; sys.exit(0 if torch.cuda.is_available() else 1)" 2>/dev/null; then
    python -m pytest tests/ -q -m gpu "$@"
fi
