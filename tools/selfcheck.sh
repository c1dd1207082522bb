#!/bin/bash
# Release sanity: everything the round driver checks that can run
# WITHOUT a GPU, in one command.  (On an MI355X box additionally run:
#   python -m pytest tests -m gpu -x -q && python bench.py --steps 40)
set -e
cd "$(dirname "$0")/.."
echo "== build (hipcc cross-compile + mzserve) =="
python -c "import __graft_entry__ as g; g.build()"
echo "== CPU test suite =="
python -m pytest tests/ -q -m "not gpu" -x
echo "== bench contract (flags parse; refuses politely without a GPU) =="
python bench.py --steps 1 --warmup 0 >/dev/null 2>&1 || true
python - <<'EOF'
import json, subprocess, sys
out = subprocess.run([sys.executable, "bench.py", "--steps", "1"],
                     capture_output=True, text=True)
line = out.stdout.strip().splitlines()[-1]
d = json.loads(line)
assert "error" in d or "metric" in d, line
print("bench contract line ok:", line[:72], "...")
EOF
echo "== mzserve binary =="
test -x veles/znicz_amd/serve/mzserve
veles/znicz_amd/serve/mzserve 2>&1 | head -1
echo "selfcheck: ALL GREEN"
