#!/bin/bash
# One-command 1/2/4/8-GPU scaling curve on a single node.
# Produces the BASELINE.json metric at each N; efficiency % = value(N) /
# (N * value(1)).  Usage: ./scripts/scale_curve.sh [extra bench.py args...]
set -u
cd "$(dirname "$0")/.."
PORT=${MASTER_PORT:-29517}
OUT=${SCALE_OUT:-gpurun_out/scale_curve.jsonl}
mkdir -p "$(dirname "$OUT")"
: > "$OUT"
AVAIL=$(python -c 'import torch; print(torch.cuda.device_count())')
for N in 1 2 4 8; do
  if [ "$N" -gt "$AVAIL" ]; then
    echo "skipping N=$N (only $AVAIL GPUs visible)"
    continue
  fi
  echo "=== N=$N ==="
  if [ "$N" -eq 1 ]; then
    python bench.py --gpus 1 --steps 20 --warmup 5 "$@" | tee -a "$OUT"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port "$PORT" \
      bench.py --gpus "$N" --steps 20 --warmup 5 "$@" | tee -a "$OUT"
  fi
done
python - "$OUT" <<'EOF'
import json, sys
rows = [json.loads(l) for l in open(sys.argv[1]) if l.strip().startswith("{")]
if rows:
    base = next((r["value"] for r in rows if r["n_gpus"] == 1), None)
    print(f"{'N':>2} {'samples/s':>12} {'eff%':>6}  rank_skew")
    for r in rows:
        eff = 100.0 * r["value"] / (r["n_gpus"] * base) if base else float("nan")
        ts = r.get("rank_times_s", [])
        skew = (max(ts) - min(ts)) / max(ts) * 100 if len(ts) > 1 else 0.0
        print(f"{r['n_gpus']:>2} {r['value']:>12.0f} {eff:>6.1f}  {skew:.1f}%")
EOF
