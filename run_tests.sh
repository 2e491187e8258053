#!/bin/bash
# Mirrors the reference's scripts/run_tests.sh: CPU suite here, GPU suite on
# an MI355X box.
set -e
python -m pytest tests -q -m "not gpu"
if python -c "import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  python -m pytest tests -q -m gpu
fi
