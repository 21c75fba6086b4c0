#!/bin/bash
# GPU round-2 call A: gpu tests + smoke + 1-proc bench + 1/2/4/8 scaling sweep
set -x
mkdir -p gpurun_out
nproc > gpurun_out/nproc.txt
cat /sys/fs/cgroup/cpu.max /sys/fs/cgroup/cpu/cpu.cfs_quota_us /sys/fs/cgroup/cpu/cpu.cfs_period_us 2>/dev/null >> gpurun_out/nproc.txt
rocm-smi --showid >> gpurun_out/nproc.txt 2>&1

timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/gputests.log 2>&1
echo "gputests rc=$?" >> gpurun_out/gputests.log

timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke.log 2>&1
echo "smoke rc=$?" >> gpurun_out/smoke.log

timeout 600 python bench.py --gpus 1 --steps 50 --warmup 10 > gpurun_out/bench_n1.json 2> gpurun_out/bench_n1.log
for N in 2 4 8; do
  timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
    --master-addr 127.0.0.1 --master-port 29513 bench.py --gpus $N --steps 50 --warmup 10 \
    > gpurun_out/bench_n$N.json 2> gpurun_out/bench_n$N.log
done
tail -1 gpurun_out/bench_n1.json gpurun_out/bench_n2.json gpurun_out/bench_n4.json gpurun_out/bench_n8.json
tail -5 gpurun_out/gputests.log gpurun_out/smoke.log
