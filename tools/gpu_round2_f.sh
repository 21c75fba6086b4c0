#!/bin/bash
# GPU round-2 call F: config-3 fleet demo with the real rocm-smi gate +
# 10k-znode scaling points.
set -x
mkdir -p gpurun_out/f
timeout 300 python tools/fleet_gpu_demo.py --seconds 20 \
  > gpurun_out/f/fleet_gpu_demo.json 2> gpurun_out/f/fleet_gpu_demo.log
echo "fleet rc=$?" >> gpurun_out/f/fleet_gpu_demo.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 4 --znodes 10000 \
  > gpurun_out/f/n1_10k.json 2> gpurun_out/f/n1_10k.log
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
  --master-addr 127.0.0.1 --master-port 29521 bench.py --gpus 8 --steps 20 --warmup 4 --znodes 10000 \
  > gpurun_out/f/n8_10k.json 2> gpurun_out/f/n8_10k.log
