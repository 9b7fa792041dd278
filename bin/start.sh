#!/bin/bash
# Entry script for the scheduler container (reference analog:
# bin/hivedscheduler/start.sh). CONFIG points at the cluster YAML; the
# process exits 0 on config change and the StatefulSet restarts it
# (work-preserving reconfiguration).
set -euo pipefail

CONFIG=${CONFIG:-/etc/hivedscheduler/config.yaml}
echo "hivedscheduler-amd starting with CONFIG=${CONFIG}"
exec python -m hivedscheduler_amd "$@"
