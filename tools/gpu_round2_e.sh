#!/bin/bash
# GPU round-2 call E: preflight with final code (ACL-skip) — gpu tests,
# smoke, n1 x3, n2 x3 (chasing the one bad n2 sample), n4, n8.
set -x
mkdir -p gpurun_out/e
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/e/gputests.log 2>&1
echo "gputests rc=$?" >> gpurun_out/e/gputests.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/e/smoke.log 2>&1
echo "smoke rc=$?" >> gpurun_out/e/smoke.log
for i in 1 2 3; do
  timeout 200 python bench.py --gpus 1 --steps 60 --warmup 10 \
    > gpurun_out/e/n1_rep$i.json 2> gpurun_out/e/n1_rep$i.log
done
for i in 1 2 3; do
  timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 2951$i bench.py --gpus 2 --steps 60 --warmup 10 \
    > gpurun_out/e/n2_rep$i.json 2> gpurun_out/e/n2_rep$i.log
done
for N in 4 8; do
  timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
    --master-addr 127.0.0.1 --master-port 29520 bench.py --gpus $N --steps 60 --warmup 10 \
    > gpurun_out/e/n$N.json 2> gpurun_out/e/n$N.log
done
