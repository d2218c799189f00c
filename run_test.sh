#!/bin/bash
# Test entry point (role of the reference's run_test.sh): CPU suite here,
# GPU suite when ROCm hardware is visible.
set -e
python -m pytest tests -q -m "not gpu" "$@"
if python -c "import torch, sys; sys.exit(0 if torch.cuda.is_available() else 1)" 2>/dev/null; then
    python -m pytest tests -q -m gpu "$@"
fi
