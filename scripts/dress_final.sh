#!/bin/bash
# Final round-2 dress rehearsal on a fresh MI355X box:
# full GPU test suite + smoke + bench at the driver's default-ish shapes.
set -u
ulimit -c 0
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
export PYTHONPATH=.

echo "== pytest -m gpu =="
timeout 800 python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee gpurun_out/dress_pytest.log

echo "== smoke =="
timeout 240 python -c 'import __graft_entry__ as g; g.build(); g.smoke(); print("SMOKE OK")' 2>&1 | tail -3 | tee gpurun_out/dress_smoke.log

echo "== bench 20/5 =="
timeout 300 python bench.py --steps 20 --warmup 5 > gpurun_out/dress_b20.json 2> gpurun_out/dress_b20.err
tail -1 gpurun_out/dress_b20.json

echo "== bench 40/6 =="
timeout 300 python bench.py --steps 40 --warmup 6 > gpurun_out/dress_b40.json 2> gpurun_out/dress_b40.err
tail -1 gpurun_out/dress_b40.json
echo "DRESS DONE"
