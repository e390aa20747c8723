#!/bin/bash
# RCCL xGMI tuning sweep — turnkey evidence generator for an 8-GPU node.
#
# Runs the flagship bench at each (NCCL_MIN_NCHANNELS, NCCL_MAX_NCHANNELS)
# point and emits profiles/rccl_channel_sweep.csv with the headline busbw
# and the 1M/16M/512M sweep points, so every env line in
# deploy/rccl/rccl-config.yaml can be justified by a measured delta
# (VERDICT r01 #2).  Usage (on a node with N GPUs):
#     bash tools/rccl_sweep.sh [NGPUS] [STEPS]
# The gpurun pool only leases 1 GPU at a time (profiles/pool_probe_r02.log)
# where this sweep is degenerate (no collective traffic); it is staged for
# the first environment with an 8-GPU lease.
set -u
NGPUS="${1:-8}"
STEPS="${2:-20}"
OUT="${OUT:-profiles/rccl_channel_sweep.csv}"
mkdir -p "$(dirname "$OUT")"
echo "min_nchannels,max_nchannels,ngpus,busbw_GBps,busbw_1M,busbw_16M,busbw_512M" > "$OUT"

run_point() {
  local mn="$1" mx="$2"
  local env_args=()
  [[ -n "$mn" ]] && export NCCL_MIN_NCHANNELS="$mn" || unset NCCL_MIN_NCHANNELS
  [[ -n "$mx" ]] && export NCCL_MAX_NCHANNELS="$mx" || unset NCCL_MAX_NCHANNELS
  local log
  log=$(mktemp)
  if [[ "$NGPUS" -gt 1 ]]; then
    timeout 600 python3 -m torch.distributed.run --nnodes=1 \
      --nproc-per-node "$NGPUS" --master-addr 127.0.0.1 --master-port 29591 \
      bench.py --gpus "$NGPUS" --steps "$STEPS" --warmup 5 > "$log" 2>&1
  else
    timeout 600 python3 bench.py --gpus 1 --steps "$STEPS" --warmup 5 \
      > "$log" 2>&1
  fi
  python3 - "$log" "$OUT" "${mn:-default}" "${mx:-default}" << 'PYEOF'
import json, sys
log, out, mn, mx = sys.argv[1:5]
line = next((l for l in open(log) if l.startswith('{"metric"')), None)
if line is None:
    print(f"sweep point min={mn} max={mx}: bench failed", file=sys.stderr)
    sys.exit(0)
r = json.loads(line)
by = {p["bytes"]: p["busbw_GBps"] for p in r["config"].get("sweep", [])}
with open(out, "a") as f:
    f.write(f'{mn},{mx},{r["n_gpus"]},{r["value"]},'
            f'{by.get(1048576,"")},{by.get(16777216,"")},'
            f'{by.get(536870912,"")}\n')
PYEOF
  rm -f "$log"
}

export HSA_ENABLE_IPC_MODE_LEGACY=0
# library defaults first (the control row), then the recipe point, then
# the neighborhood: per-link rings on 7 xGMI links suggest multiples of 7,
# per-direction 14; XCD-aligned 8/16/32; the recipe's 28; and wide 56/64.
run_point "" ""
for mn in 8 14 16 28 32 56; do
  run_point "$mn" 64
done
run_point 28 28
echo "sweep complete -> $OUT"
cat "$OUT"
