#!/bin/bash
# Sharded-controller scaling check on the 256-core MI355X node (fake SMI for
# multi-rank: the box has 1 GPU; the control plane is the thing under test).
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
  --master-addr 127.0.0.1 --master-port 29541 bench.py --gpus 4 --steps 300 --warmup 30 --fake \
  > gpurun_out/bench6_w4.json 2> gpurun_out/bench6_w4.log
echo "w4 rc=$?"
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
  --master-addr 127.0.0.1 --master-port 29542 bench.py --gpus 8 --steps 300 --warmup 30 --fake \
  > gpurun_out/bench6_w8.json 2> gpurun_out/bench6_w8.log
echo "w8 rc=$?"
# single-rank real backend: confirm no regression from the shard plumbing
timeout 400 python bench.py --steps 400 --warmup 40 > gpurun_out/bench6_1.json 2> gpurun_out/bench6_1.log
echo "b1 rc=$?"
