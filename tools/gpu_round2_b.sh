#!/bin/bash
# Round-2 GPU call B:
#  1. RCCL env-uptake proof: run the native harness with the rccl-config
#     recipe exported and NCCL_DEBUG=INFO; RCCL logs every env knob it
#     parses ("... set by environment to ...") — evidence the ConfigMap
#     recipe actually takes effect in this image's librccl.
#  2. Fresh rocprofv3 kernel-stats profile of the round-2 bench.
set -u
mkdir -p gpurun_out
cd /root/repo

echo "=== 1. RCCL env uptake (NCCL_DEBUG=INFO) ==="
export HSA_ENABLE_IPC_MODE_LEGACY=0
export NCCL_MIN_NCHANNELS=28
export NCCL_MAX_NCHANNELS=64
export NCCL_DEBUG=INFO
timeout 300 ./cea_amd/bin/all_reduce_perf -b 1M -e 16M -f 2 -g 1 -w 2 -n 10 -c 0 \
  > gpurun_out/r02b_env_uptake.log 2>&1
rc=$?
echo "harness rc=$rc"
grep -i "set by environment\|NCHANNELS" gpurun_out/r02b_env_uptake.log | head -20
unset NCCL_DEBUG NCCL_MIN_NCHANNELS NCCL_MAX_NCHANNELS

echo "=== 2. rocprofv3 kernel stats of bench ==="
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/r02b_prof -o r02b -- \
  python3 bench.py --gpus 1 --steps 20 --warmup 5 --no-sweep \
  > gpurun_out/r02b_bench.log 2>&1
echo "rocprof rc=$?"
find gpurun_out/r02b_prof -name "*stats*" | head -3
for f in $(find gpurun_out/r02b_prof -name "*kernel_stats*" | head -1); do
  head -8 "$f"
done
echo "=== done ==="
