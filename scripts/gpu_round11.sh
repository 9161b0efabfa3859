#!/bin/bash
# Full validation after reconnect/persistence/wake-deferral work.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python build_native.py > gpurun_out/build11.log 2>&1
echo "build rc=$?"
timeout 900 python -m pytest tests -x -q -m "not gpu" > gpurun_out/pytest_cpu11.log 2>&1
echo "pytest-cpu rc=$?"
timeout 300 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu11.log 2>&1
echo "pytest-gpu rc=$?"
timeout 400 python bench.py --steps 500 --warmup 50 > gpurun_out/bench11_1.json 2> gpurun_out/bench11_1.log
echo "b1 rc=$?"
for W in 2 4 8; do
  timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node $W \
    --master-addr 127.0.0.1 --master-port 2958$W bench.py --gpus $W --steps 400 --warmup 40 --fake \
    > gpurun_out/bench11_w$W.json 2> gpurun_out/bench11_w$W.log
  echo "w$W rc=$?"
done
timeout 600 python -m benchmarks.scenarios --scenario all > gpurun_out/scenarios11.json 2> gpurun_out/scenarios11.log
echo "scenarios rc=$?"
