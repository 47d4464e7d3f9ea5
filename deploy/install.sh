#!/usr/bin/env bash
# gpud-amd installer (reference: install.sh — fetch + install + systemd).
# Installs the package into /opt/gpud-amd, builds the native extensions for
# gfx950, and registers the systemd unit.
set -euo pipefail

PREFIX=${PREFIX:-/opt/gpud-amd}
DATA_DIR=${DATA_DIR:-/var/lib/gpud}
SRC_DIR=$(cd "$(dirname "$0")/.." && pwd)

if [ "$(id -u)" != 0 ]; then
  echo "install.sh must run as root" >&2
  exit 1
fi

command -v hipcc >/dev/null || { echo "ROCm (hipcc) is required" >&2; exit 1; }
command -v python3 >/dev/null || { echo "python3 is required" >&2; exit 1; }

echo "installing to ${PREFIX}"
mkdir -p "${PREFIX}" "${DATA_DIR}"
cp -r "${SRC_DIR}/gpud_amd" "${SRC_DIR}/csrc" "${PREFIX}/"
(cd "${PREFIX}" && bash csrc/build.sh)

PYTHONPATH="${PREFIX}" python3 -m gpud_amd up --data-dir "${DATA_DIR}" "$@"
echo "gpud-amd installed; check: systemctl status gpud-amd"
