#!/usr/bin/env bash
# CPU test suite (parity with reference run_ci_tests.sh).
set -euo pipefail
cd "$(dirname "$0")"
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
python -m pytest tests -q -m "not gpu"
