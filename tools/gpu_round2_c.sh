#!/bin/bash
# GPU round-2 call C: endurance revalidation of the r2 concurrency changes
# (multi stat snapshot, NodeDeleted union watches, per-op zxid, decoder
# hardening) + native chaos soak + a 10k-znode latency sweep.
set -x
mkdir -p gpurun_out
timeout 360 python tools/endurance.py --seconds 240 --znodes 500 \
  > gpurun_out/endurance_r2_default.json 2> gpurun_out/endurance_r2_default.log
echo "endurance default rc=$?" >> gpurun_out/endurance_r2_default.log
timeout 360 python tools/endurance.py --seconds 240 --znodes 1000 --atomic --gpu \
  > gpurun_out/endurance_r2_atomic_gpu.json 2> gpurun_out/endurance_r2_atomic_gpu.log
echo "endurance atomic rc=$?" >> gpurun_out/endurance_r2_atomic_gpu.log
timeout 120 ./bin/stress -c 8 -t 30 > gpurun_out/soak_r2.txt 2>&1
echo "soak rc=$?" >> gpurun_out/soak_r2.txt
timeout 300 python tools/latency_sweep.py > gpurun_out/latency_sweep_r2.json 2> gpurun_out/latency_sweep_r2.log
echo "latency rc=$?" >> gpurun_out/latency_sweep_r2.log
