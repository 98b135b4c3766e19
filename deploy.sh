#!/usr/bin/env bash
# Build and install the package for a node deployment: compile the in-tree
# gfx950 HIP extensions, build an sdist, install locally, optionally copy
# the artifact to a target host/dir (DEPLOY_TARGET). Counterpart of the
# reference's deploy.sh (build sdist -> pip install -> ship to simulator).
set -euo pipefail
cd "$(dirname "$0")"

python -c "from __graft_entry__ import build; build()"   # hipcc, in-tree .so
python setup.py sdist

if [ -n "${DEPLOY_TARGET:-}" ]; then
    scp dist/*.tar.gz "$DEPLOY_TARGET" 2>/dev/null || cp dist/*.tar.gz "$DEPLOY_TARGET"
    echo "shipped $(ls dist/*.tar.gz | tail -1) -> $DEPLOY_TARGET"
fi
