#!/bin/bash
# CI entry: build everything, run the CPU suite, sanitize the C++ runtime.
# (GPU tier — pytest -m gpu + bench — runs on an MI355X box.)
set -euo pipefail
cd "$(dirname "$0")/.."
python -c "import __graft_entry__ as g; g.build()"
python -m pytest tests -q -m "not gpu"
scripts/sanitize_check.sh address
scripts/sanitize_check.sh thread
echo "CI OK"
