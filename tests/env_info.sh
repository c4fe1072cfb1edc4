#!/usr/bin/env bash
# Environment dump (parity with reference tests/env_info.sh).
set +e
uname -a
python --version
python -c "import torch; print('torch', torch.__version__, 'cuda?', torch.cuda.is_available())"
python -c "import pyarrow; print('pyarrow', pyarrow.__version__)"
/opt/rocm/bin/hipcc --version 2>/dev/null | head -2
rocm-smi 2>/dev/null | head -10
