#!/usr/bin/env bash
# CI pipeline matrix (the reference runs 15 drone pipelines incl. a
# gpucloud hardware matrix, .drone.yml:1177; this is the equivalent
# staged gate for one repo):
#
#   ./scripts/ci.sh [stage]
#
# stages: lint build unit leaks gpu gpuvalidate bench all (default: all
# CPU stages, plus GPU stages when a GPU is visible)
set -euo pipefail
cd "$(dirname "$0")/.."
STAGE="${1:-auto}"

run_lint() {
  echo "=== lint: compile-check every module ==="
  python -m compileall -q helix_amd tests scripts bench.py
}

run_build() {
  echo "=== build: gfx950 extension (hipcc cross-compile) ==="
  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
}

run_unit() {
  echo "=== unit: CPU suite (incl. gloo multi-process) ==="
  python -m pytest tests -x -q -m "not gpu" --timeout 600
}

run_leaks() {
  echo "=== leaks: thread/process hygiene (race-detector role) ==="
  # every test runs under pytest-timeout; a leaked thread or child
  # process hangs teardown and trips it — plus an explicit sweep:
  python - <<'PY'
import subprocess, sys
out = subprocess.run(
    [sys.executable, "-m", "pytest",
     "tests/test_store_concurrency.py", "tests/test_tp_instance.py",
     "-q", "--timeout", "300"], capture_output=True, text=True)
print(out.stdout[-2000:])
sys.exit(out.returncode)
PY
}

run_gpu() {
  echo "=== gpu: numerics + engine suite on hardware ==="
  python -m pytest tests -x -q -m gpu --timeout 900
}

run_gpuvalidate() {
  echo "=== gpuvalidate: runner-plane scenarios (gpucloud parity) ==="
  timeout 700 python scripts/gpu_validate.py --small
}

run_bench() {
  echo "=== bench: headline sanity ==="
  python bench.py --steps 8 --warmup 4
}

HAS_GPU=0
python -c "import torch,sys;sys.exit(0 if torch.cuda.is_available() else 1)" \
  && HAS_GPU=1 || true

case "$STAGE" in
  lint) run_lint ;;
  build) run_build ;;
  unit) run_unit ;;
  leaks) run_leaks ;;
  gpu) run_gpu ;;
  gpuvalidate) run_gpuvalidate ;;
  bench) run_bench ;;
  auto|all)
    run_lint; run_build; run_unit; run_leaks
    if [ "$HAS_GPU" = 1 ]; then
      run_gpu; run_gpuvalidate; run_bench
    else
      echo "(no GPU visible: gpu/gpuvalidate/bench stages skipped)"
    fi
    ;;
  *) echo "unknown stage: $STAGE"; exit 2 ;;
esac
echo "CI OK ($STAGE)"
