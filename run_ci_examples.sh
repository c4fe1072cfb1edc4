#!/usr/bin/env bash
# Example smoke runs (parity with reference run_ci_examples.sh:8-9: the
# dataset/torch_dataset __main__ demos, 1e6 rows over 10 files, 4 epochs).
set -euo pipefail
cd "$(dirname "$0")"
python -m ray_shuffling_data_loader_amd.dataset
python -m ray_shuffling_data_loader_amd.torch_dataset
