#!/bin/bash
# GPU round-2 call I (budget capstone): back-to-back single-daemon and fleet
# endurance on the final tree, spending the remaining round budget on
# robustness evidence.
set -x
mkdir -p gpurun_out/i
timeout 600 python tools/endurance.py --seconds 500 --znodes 1000 --atomic --gpu \
  > gpurun_out/i/endurance_single.json 2> gpurun_out/i/endurance_single.log
echo "single rc=$?" >> gpurun_out/i/endurance_single.log
timeout 600 python tools/fleet_endurance.py --seconds 500 --gpu \
  > gpurun_out/i/endurance_fleet.json 2> gpurun_out/i/endurance_fleet.log
echo "fleet rc=$?" >> gpurun_out/i/endurance_fleet.log
