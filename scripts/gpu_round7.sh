#!/bin/bash
# Native-store + sharded-controller validation on the 256-core MI355X node.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python build_native.py > gpurun_out/build7.log 2>&1
echo "build rc=$?"
timeout 600 python -m pytest tests -x -q -m "not gpu" > gpurun_out/pytest_cpu7.log 2>&1
echo "pytest-cpu rc=$?"
timeout 300 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu7.log 2>&1
echo "pytest-gpu rc=$?"
# 1-rank real backend
timeout 400 python bench.py --steps 500 --warmup 50 > gpurun_out/bench7_1.json 2> gpurun_out/bench7_1.log
echo "b1 rc=$?"
# scaling matrix (fake SMI: the box has 1 GPU; control plane is under test)
for W in 2 4 8; do
  timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node $W \
    --master-addr 127.0.0.1 --master-port 2955$W bench.py --gpus $W --steps 300 --warmup 30 --fake \
    > gpurun_out/bench7_w$W.json 2> gpurun_out/bench7_w$W.log
  echo "w$W rc=$?"
done
