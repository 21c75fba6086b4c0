#!/bin/bash
# GPU round-2 call H (final): preflight on the final tree + 10-min fleet
# endurance with the real gpu-liveness gate.
set -x
mkdir -p gpurun_out/h
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/h/gputests.log 2>&1
echo "gputests rc=$?" >> gpurun_out/h/gputests.log
timeout 300 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/h/build.log 2>&1
echo "build rc=$?" >> gpurun_out/h/build.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/h/smoke.log 2>&1
echo "smoke rc=$?" >> gpurun_out/h/smoke.log
timeout 200 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/h/n1.json 2> gpurun_out/h/n1.log
for N in 2 4 8; do
  timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
    --master-addr 127.0.0.1 --master-port 29530 bench.py --gpus $N --steps 60 --warmup 10 \
    > gpurun_out/h/n$N.json 2> gpurun_out/h/n$N.log
done
timeout 700 python tools/fleet_endurance.py --seconds 600 --gpu \
  > gpurun_out/h/fleet_endurance_gpu.json 2> gpurun_out/h/fleet_endurance_gpu.log
echo "fleet endurance rc=$?" >> gpurun_out/h/fleet_endurance_gpu.log
