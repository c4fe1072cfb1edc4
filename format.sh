#!/usr/bin/env bash
# Lint (parity with reference format.sh). flake8 if available.
set -euo pipefail
cd "$(dirname "$0")"
python -m flake8 --max-line-length 100 ray_shuffling_data_loader_amd tests bench.py 2>/dev/null \
  || python -m pyflakes ray_shuffling_data_loader_amd 2>/dev/null \
  || echo "no linter available; skipped"
