#!/bin/bash
# One-shot 1/2/4/8-GPU scaling curve on a single MI355X node (VERDICT r01
# item 7).  The round-end driver produces SCALE_rNN.json itself by running
# bench.py at N=1,2,4,8; this script emits the same curve for manual runs.
#
# RCCL over xGMI notes baked into the defaults:
#  - HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC; required on this host stack)
#  - bucket size 64 MB (bench.py --bucket-mb; ring allreduce on 7 p2p links
#    is per-link bound -> few large buckets, see parallel/ddp.py)
set -u
STEPS=${STEPS:-16}
WARMUP=${WARMUP:-4}
OUT=${OUT:-scale_curve.jsonl}
: > "$OUT"
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
for N in 1 2 4 8; do
  echo "=== dp$N ===" >&2
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port "${PORT:-29571}" \
    bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP" \
    2>/dev/null | tail -1 | tee -a "$OUT"
done
python - "$OUT" <<'PY'
import json, sys
rows = [json.loads(l) for l in open(sys.argv[1]) if l.strip()]
base = next((r for r in rows if r['n_gpus'] == 1), None)
if base:
    for r in rows:
        eff = r['value'] / (base['value'] * r['n_gpus'])
        print(f"dp{r['n_gpus']}: {r['value']:.0f} {r['unit']}  "
              f"(weak-scaling efficiency {eff:.3f})")
PY
