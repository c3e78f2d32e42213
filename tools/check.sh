#!/bin/bash
# CI-style gate: build (fail loudly — never pipe the build through tail
# without pipefail; a swallowed hipcc failure once shipped a stale .so to
# a GPU validation run) + full CPU suite.
set -euo pipefail
cd "$(dirname "$0")/.."
python -c "from __graft_entry__ import build; build()"
python -m pytest tests/ -q -m "not gpu"
echo "[check] OK"
