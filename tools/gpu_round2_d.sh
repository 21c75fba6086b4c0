#!/bin/bash
# GPU round-2 call D: n1 repeat stats + final sweep + 12-min endurance capstone
set -x
mkdir -p gpurun_out/d
for i in 1 2 3 4 5; do
  timeout 200 python bench.py --gpus 1 --steps 60 --warmup 10 \
    > gpurun_out/d/n1_rep$i.json 2> gpurun_out/d/n1_rep$i.log
done
for N in 2 4 8; do
  timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
    --master-addr 127.0.0.1 --master-port 29515 bench.py --gpus $N --steps 60 --warmup 10 \
    > gpurun_out/d/n$N.json 2> gpurun_out/d/n$N.log
done
timeout 800 python tools/endurance.py --seconds 720 --znodes 1000 --atomic --gpu \
  > gpurun_out/d/endurance_capstone_r2.json 2> gpurun_out/d/endurance_capstone_r2.log
echo "endurance rc=$?" >> gpurun_out/d/endurance_capstone_r2.log
