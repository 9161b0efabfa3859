#!/bin/bash
# Round-1 session 2: validate the latency-overhauled control plane on MI355X.
set -x
cd /root/repo
mkdir -p gpurun_out
nproc > gpurun_out/nproc.txt
# 1) native build check + gpu test tier
timeout 300 python build_native.py > gpurun_out/build5.log 2>&1
echo "build rc=$?"
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu5.log 2>&1
echo "pytest rc=$?"
# 2) flagship bench, real amdsmi backend
timeout 500 python bench.py --steps 400 --warmup 40 > gpurun_out/bench5.json 2> gpurun_out/bench5.log
echo "bench rc=$?"
# 3) control-plane scaling preview on the box's faster cores (fake SMI, 4+8 ranks)
timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
  --master-addr 127.0.0.1 --master-port 29531 bench.py --gpus 4 --steps 200 --warmup 20 --fake \
  > gpurun_out/bench5_w4.json 2> gpurun_out/bench5_w4.log
echo "w4 rc=$?"
timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
  --master-addr 127.0.0.1 --master-port 29532 bench.py --gpus 8 --steps 200 --warmup 20 --fake \
  > gpurun_out/bench5_w8.json 2> gpurun_out/bench5_w8.log
echo "w8 rc=$?"
