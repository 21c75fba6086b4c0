#!/bin/bash
# GPU round-2 call B: CPU-split experiment matrix for the scaling bench.
# Strategies per N: default (2cpu/rank), rank1 (BENCH_RANK_CPUS=1),
# rank3 (3/rank where it fits), none (BENCH_AFFINITY=0). Short runs.
set -x
mkdir -p gpurun_out/expt
run() {  # run N tag [env...]
  local N=$1 tag=$2; shift 2
  if [ "$N" = 1 ]; then
    env "$@" timeout 300 python bench.py --gpus 1 --steps 40 --warmup 8 \
      > gpurun_out/expt/n${N}_${tag}.json 2> gpurun_out/expt/n${N}_${tag}.log
  else
    env "$@" timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29514 bench.py --gpus $N --steps 40 --warmup 8 \
      > gpurun_out/expt/n${N}_${tag}.json 2> gpurun_out/expt/n${N}_${tag}.log
  fi
}
for N in 2 4 8; do
  run $N default
  run $N rank1 BENCH_RANK_CPUS=1
  run $N none BENCH_AFFINITY=0
done
run 4 rank3 BENCH_RANK_CPUS=3
run 2 rank4 BENCH_RANK_CPUS=4
run 1 default
run 1 split4 BENCH_RANK_CPUS=4
run 1 split8 BENCH_RANK_CPUS=8
grep -h -o '"n_gpus": [0-9]*, .*"ms_per_step": [0-9.]*' gpurun_out/expt/*.json | head -40 || true
