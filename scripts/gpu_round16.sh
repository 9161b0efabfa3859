#!/bin/bash
# Comprehensive matrix after events/preemption/gang work.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python build_native.py > gpurun_out/build16.log 2>&1
echo "build rc=$?"
timeout 900 python -m pytest tests -q -m "not gpu" > gpurun_out/pytest_cpu16.log 2>&1
echo "cpu rc=$?"
timeout 300 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu16.log 2>&1
echo "gpu rc=$?"
timeout 400 python bench.py --steps 600 --warmup 60 > gpurun_out/bench16_1.json 2> gpurun_out/bench16_1.log
echo "b1 rc=$?"
for W in 2 4 8; do
  timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node $W \
    --master-addr 127.0.0.1 --master-port 2960$W bench.py --gpus $W --steps 400 --warmup 40 --fake \
    > gpurun_out/bench16_w$W.json 2> gpurun_out/bench16_w$W.log
  echo "w$W rc=$?"
done
timeout 900 python -m benchmarks.scenarios --scenario all > gpurun_out/scenarios16.json 2> gpurun_out/scenarios16.log
echo "scenarios rc=$?"
