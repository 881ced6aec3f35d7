#!/usr/bin/env bash
# Shape-compatible launcher for the AMD CC manager (the reference ships
# a scripts/cc-manager.sh-style entrypoint; container ENTRYPOINT mirrors
# /root/reference/deployments/container/Dockerfile.distroless:81).
set -euo pipefail
exec python3 -m k8s_cc_manager_amd "$@"
