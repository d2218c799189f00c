#!/usr/bin/env bash
# One-shot CPU-side validation battery (everything that runs without a GPU):
#   ./scripts/validate.sh          # suite + all sanitizer/mock harnesses
#   ./scripts/validate.sh quick    # suite only
# GPU-side equivalents (run on an MI355X box):
#   python -m pytest tests -q -m gpu
#   python bench.py --steps 30 --warmup 3
#   python scripts/soak.py --seconds 60 --threads 6 --quant-frac 0.4
set -e
cd "$(dirname "$0")/.."

echo "== CPU test suite =="
python -m pytest tests -q -m "not gpu"

[ "$1" = "quick" ] && exit 0

for mode in asan tsan mockverbs mockverbs-asan mockverbs-tsan; do
    echo "== $mode =="
    python scripts/san_build.py "$mode"
done
echo "ALL GREEN"
