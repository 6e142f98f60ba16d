#!/bin/bash
# Consolidated end-of-round-2 validation on a fresh MI355X box.
set -u
ulimit -c 0
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
export PYTHONPATH=.

echo "== pytest -m gpu (95) =="
timeout 800 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee gpurun_out/final_pytest.log

echo "== smoke =="
timeout 240 python -c 'import __graft_entry__ as g; g.build(); g.smoke(); print("SMOKE OK")' 2>&1 | tail -2

echo "== bench 20/5 x3 =="
for i in 1 2 3; do
  timeout 200 python bench.py --steps 20 --warmup 5 2>/dev/null | tail -1 \
    | python -c 'import json,sys; d=json.loads(sys.stdin.read()); print("run", d["value"], d["ms_per_step"], d["config"]["jobs_completed"])'
done | tee gpurun_out/final_bench_spread.txt
timeout 200 python bench.py --steps 40 --warmup 6 2>/dev/null | tail -1 > gpurun_out/final_b40.json
tail -1 gpurun_out/final_b40.json

echo "== gandiva soak (calibrated + lib mode) =="
timeout 300 python scripts/soak_gandiva.py 240 2>&1 | tail -1
mv gpurun_out/soak_r2.jsonl gpurun_out/soak_final.jsonl 2>/dev/null

echo "== attn NW A/B at llama shape =="
for nw in 4 8; do
  GSAMD_ATTN_NW=$nw timeout 120 python - <<'EOF'
import os, time, torch
from gpuschedule_amd.ops.attention import flash_attention
q = torch.randn(1, 32, 2048, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(1, 8, 2048, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn_like(k)
for _ in range(10): flash_attention(q, k, v, causal=True)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(50): flash_attention(q, k, v, causal=True)
torch.cuda.synchronize()
print(f"NW={os.environ['GSAMD_ATTN_NW']}: "
      f"{(time.perf_counter()-t0)/50*1e3:.3f} ms")
EOF
done 2>&1 | grep "NW=" | tee gpurun_out/final_attn_nw.txt
echo "FINAL DONE"
