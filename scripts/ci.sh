#!/usr/bin/env bash
# CI entrypoint (the reference runs drone/gha pipelines; this is the
# equivalent gate): build the gfx950 extension, run the CPU suite,
# then — if a GPU is visible — the GPU suite and a short bench.
set -euo pipefail
cd "$(dirname "$0")/.."
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
python -m pytest tests -x -q -m "not gpu"
if python -c "import torch, sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  python -m pytest tests -x -q -m gpu
  python bench.py --steps 4 --warmup 2
fi
