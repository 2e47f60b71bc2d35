#!/bin/bash
# End-of-round GPU validation: full gpu suite + smoke. Run via gpurun.
set -o pipefail
cd "$(dirname "$0")/.."
python -m pytest tests -q -m gpu --timeout 600 2>&1 | tail -6
echo "===SMOKE==="
python - <<'PY'
import __graft_entry__ as g
g.smoke()
print("smoke-ok")
PY
