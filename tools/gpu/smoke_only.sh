#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
timeout 120 python -c "import __graft_entry__; __graft_entry__.smoke()"
