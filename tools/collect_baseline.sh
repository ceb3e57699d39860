#!/usr/bin/env bash
# Regenerate every BASELINE.md measurement on an MI355X box.
# Usage (on a GPU box, repo root):  bash tools/collect_baseline.sh [outdir]
# Or through gpurun:
#   gpurun --timeout 3000 -- 'bash tools/collect_baseline.sh gpurun_out/baseline'
set -uo pipefail
OUT=${1:-gpurun_out/baseline}
mkdir -p "$OUT"
log() { echo "== $*" | tee -a "$OUT/summary.txt"; }

log "GPU test suite"
python -m pytest tests -m gpu -q 2>&1 | grep -E "passed|failed" | tail -1 \
  | tee -a "$OUT/summary.txt"

log "config 2: linear 10M-row cycle (steady state)"
timeout 400 python bench.py --rows 10000000 --steps 10 --warmup 3 \
  > "$OUT/bench_linear_10m.json" 2>"$OUT/bench_linear_10m.log"
tail -1 "$OUT/bench_linear_10m.json" | tee -a "$OUT/summary.txt"

log "config 3 scale/GPU: 125M rows"
timeout 700 python bench.py --rows 125000000 --steps 2 --warmup 1 \
  > "$OUT/bench_linear_125m.json" 2>/dev/null
tail -1 "$OUT/bench_linear_125m.json" | tee -a "$OUT/summary.txt"

log "config 5: MLP-4096, 1M rows"
timeout 700 python bench.py --model mlp --rows 1000000 --steps 3 --warmup 2 \
  --mlp-steps 20 > "$OUT/bench_mlp_1m.json" 2>/dev/null
tail -1 "$OUT/bench_mlp_1m.json" | tee -a "$OUT/summary.txt"

log "config 5b: MLP-4096, 10M rows"
timeout 700 python bench.py --model mlp --rows 10000000 --steps 2 --warmup 1 \
  > "$OUT/bench_mlp_10m.json" 2>/dev/null
tail -1 "$OUT/bench_mlp_10m.json" | tee -a "$OUT/summary.txt"

log "full subprocess DAG (runner, 2 serving replicas)"
python - <<'PYEOF'
from datetime import date
from bodywork_mlops_demo_amd.store import LocalStore
from bodywork_mlops_demo_amd.stages import datagen
datagen.run(LocalStore("/tmp/cb_dagstore"), n=1440, date=date(2026, 2, 1),
            device="cuda:0")
PYEOF
timeout 400 python bodywork_mlops_demo_amd/pipeline/runner.py pipeline.yaml \
  --store /tmp/cb_dagstore > "$OUT/dag.log" 2>&1
grep -E "captured scoring|pipeline run" "$OUT/dag.log" | tail -2 | tee -a "$OUT/summary.txt"

log "config 4: 30-day drift loop (reference scale)"
timeout 500 python -m bodywork_mlops_demo_amd loop --days 30 --rows 1440 \
  --store "$OUT/loopstore" --device cuda:0 \
  --json-out "$OUT/loop30.json" > "$OUT/loop30.log" 2>&1
python -m bodywork_mlops_demo_amd analytics --store "$OUT/loopstore" \
  2>/dev/null | tail -2 | tee -a "$OUT/summary.txt"

log "r2: linear 10M, GROWING read-all history"
timeout 400 python bench.py --rows 10000000 --steps 10 --warmup 2 \
  --history all > "$OUT/bench_linear_histall.json" 2>/dev/null
tail -1 "$OUT/bench_linear_histall.json" | tee -a "$OUT/summary.txt"

log "r2: linear 10M over the HTTP binary wire (uvicorn replica in-cycle)"
timeout 500 python bench.py --rows 10000000 --steps 5 --warmup 2 \
  --serving http > "$OUT/bench_linear_http.json" 2>/dev/null
tail -1 "$OUT/bench_linear_http.json" | tee -a "$OUT/summary.txt"

log "r2: MLP 125M bf16 vs MX-fp8 scoring A/B"
timeout 700 python bench.py --model mlp --rows 125000000 --steps 2 \
  --warmup 1 > "$OUT/bench_mlp_125m_bf16.json" 2>/dev/null
tail -1 "$OUT/bench_mlp_125m_bf16.json" | tee -a "$OUT/summary.txt"
timeout 700 python bench.py --model mlp-fp8 --rows 125000000 --steps 2 \
  --warmup 1 > "$OUT/bench_mlp_125m_fp8.json" 2>/dev/null
tail -1 "$OUT/bench_mlp_125m_fp8.json" | tee -a "$OUT/summary.txt"

log "r2: MX-fp8 GEMM check + A/B microbench"
timeout 500 python tools/probe_mx8.py --check --bench 2>/dev/null \
  | tee -a "$OUT/summary.txt"

log "GEMM microbench"
python - 2>/dev/null <<'EOF' | tee -a "$OUT/summary.txt"
import torch, time
from bodywork_mlops_demo_amd import ops
def bench(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n
for M, N, K in [(4096, 4096, 4096), (8192, 8192, 8192)]:
    x = (torch.randn(M, K, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.5).bfloat16()
    t = bench(lambda: ops.linear_bf16(x, w))
    print(f"NT GEMM {M}x{N}x{K}: {2*M*N*K/t/1e12:.0f} TF")
EOF

log "done; see $OUT/"
