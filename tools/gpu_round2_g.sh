#!/bin/bash
# GPU round-2 call G: fixed fleet demo (gpu payload block) + 30-min
# endurance capstone on final r2 code.
set -x
mkdir -p gpurun_out/g
timeout 300 python tools/fleet_gpu_demo.py --seconds 20 \
  > gpurun_out/g/fleet_gpu_demo.json 2> gpurun_out/g/fleet_gpu_demo.log
echo "fleet rc=$?" >> gpurun_out/g/fleet_gpu_demo.log
timeout 1900 python tools/endurance.py --seconds 1800 --znodes 1000 --atomic --gpu \
  > gpurun_out/g/endurance_capstone30_r2.json 2> gpurun_out/g/endurance_capstone30_r2.log
echo "endurance rc=$?" >> gpurun_out/g/endurance_capstone30_r2.log
