#!/bin/bash
# Round-2 GPU call A: validate round-2 changes on hardware.
#  1. pytest -m gpu (full hardware tier with the round-2 code)
#  2. bench.py quick run (driver contract intact)
#  3. dual-rank-on-1-GPU RCCL experiment (does RCCL accept 2 ranks on one
#     device? if yes we get a real collective measurement this round)
set -u
mkdir -p gpurun_out
cd /root/repo

echo "=== 1. GPU test tier ===" | tee gpurun_out/r02a_summary.txt
timeout 900 python3 -m pytest tests -m gpu -x -q 2>&1 | tail -5 | tee -a gpurun_out/r02a_summary.txt

echo "=== 2. bench quick ===" | tee -a gpurun_out/r02a_summary.txt
timeout 600 python3 bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/r02a_bench.json 2> gpurun_out/r02a_bench.err
tail -1 gpurun_out/r02a_bench.json | head -c 600 | tee -a gpurun_out/r02a_summary.txt
echo | tee -a gpurun_out/r02a_summary.txt

echo "=== 3. dual-rank single-GPU RCCL experiment ===" | tee -a gpurun_out/r02a_summary.txt
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 300 python3 -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29581 \
  bench.py --gpus 2 --steps 10 --warmup 3 --no-sweep \
  > gpurun_out/r02a_dual.log 2>&1
rc=$?
echo "dual-rank rc=$rc" | tee -a gpurun_out/r02a_summary.txt
tail -20 gpurun_out/r02a_dual.log | tee -a gpurun_out/r02a_summary.txt
echo "=== done ===" | tee -a gpurun_out/r02a_summary.txt
