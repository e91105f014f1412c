#!/usr/bin/env bash
# helix-amd installer (parity role of the reference's install.sh /
# `stack` dev driver): single-node setup on an MI355X box.
#
#   ./deploy/install.sh [--mode systemd|compose|dev] [--with-runner]
#
# - systemd: installs unit files for control plane (+ runner) and
#   starts them (default).
# - compose: docker compose up with the split-plane stack.
# - dev: foreground `serve --local-runner` for development.
set -euo pipefail

MODE=systemd
WITH_RUNNER=0
for arg in "$@"; do
  case "$arg" in
    --mode) ;;
    systemd|compose|dev) MODE="$arg" ;;
    --mode=*) MODE="${arg#--mode=}" ;;
    --with-runner) WITH_RUNNER=1 ;;
    -h|--help) grep '^#' "$0" | sed 's/^# \{0,1\}//'; exit 0 ;;
  esac
done

ROOT="$(cd "$(dirname "$0")/.." && pwd)"
cd "$ROOT"

echo "==> checking prerequisites"
python3 -c "import torch" || { echo "PyTorch-ROCm required"; exit 1; }
if [ "$WITH_RUNNER" = 1 ] || [ "$MODE" = dev ]; then
  python3 -c "import torch; assert torch.cuda.is_available(), 'no GPU'" \
    || { echo "runner requested but no GPU visible"; exit 1; }
fi

echo "==> building the gfx950 extension in-tree"
PYTORCH_ROCM_ARCH=gfx950 python3 setup.py build_ext --inplace

echo "==> generating credentials (idempotent)"
ENV_FILE=/etc/helix-amd.env
if [ ! -f "$ENV_FILE" ]; then
  ADMIN_KEY="hl-admin-$(head -c24 /dev/urandom | xxd -p)"
  RUNNER_TOKEN="hl-runner-$(head -c24 /dev/urandom | xxd -p)"
  {
    echo "HELIX_ADMIN_API_KEY=$ADMIN_KEY"
    echo "HELIX_RUNNER_TOKEN=$RUNNER_TOKEN"
    echo "HELIX_STORE_PATH=/var/lib/helix-amd/helix.db"
    echo "HELIX_FILESTORE_PATH=/var/lib/helix-amd/filestore"
    echo "HSA_ENABLE_IPC_MODE_LEGACY=0"
  } > "$ENV_FILE"
  chmod 600 "$ENV_FILE"
  echo "    wrote $ENV_FILE (admin key: $ADMIN_KEY)"
else
  echo "    $ENV_FILE exists, keeping it"
fi
mkdir -p /var/lib/helix-amd

case "$MODE" in
  systemd)
    echo "==> installing systemd units"
    sed "s|__ROOT__|$ROOT|g" deploy/helix-amd.service \
      > /etc/systemd/system/helix-amd.service
    if [ "$WITH_RUNNER" = 1 ]; then
      cat > /etc/systemd/system/helix-amd-runner.service <<UNIT
[Unit]
Description=helix-amd GPU runner
After=helix-amd.service
[Service]
EnvironmentFile=$ENV_FILE
WorkingDirectory=$ROOT
ExecStart=/usr/bin/python3 -m helix_amd.cli runner --api-url http://127.0.0.1:8080 --runner-id %H --tunnel
Restart=on-failure
[Install]
WantedBy=multi-user.target
UNIT
    fi
    systemctl daemon-reload
    systemctl enable --now helix-amd
    [ "$WITH_RUNNER" = 1 ] && systemctl enable --now helix-amd-runner
    echo "==> done: curl -H \"Authorization: Bearer \$HELIX_ADMIN_API_KEY\" http://127.0.0.1:8080/healthz"
    ;;
  compose)
    echo "==> docker compose up"
    docker compose -f deploy/docker-compose.yaml up -d --build
    ;;
  dev)
    echo "==> dev mode (foreground, local runner)"
    exec python3 -m helix_amd.cli serve --local-runner
    ;;
  *)
    echo "unknown mode: $MODE"; exit 1 ;;
esac
