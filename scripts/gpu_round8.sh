#!/bin/bash
# CAS placement + batched teardown validation; scenario suite refresh.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python build_native.py > gpurun_out/build8.log 2>&1
echo "build rc=$?"
timeout 300 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu8.log 2>&1
echo "pytest-gpu rc=$?"
timeout 400 python bench.py --steps 500 --warmup 50 > gpurun_out/bench8_1.json 2> gpurun_out/bench8_1.log
echo "b1 rc=$?"
for W in 2 4 8; do
  timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node $W \
    --master-addr 127.0.0.1 --master-port 2956$W bench.py --gpus $W --steps 300 --warmup 30 --fake \
    > gpurun_out/bench8_w$W.json 2> gpurun_out/bench8_w$W.log
  echo "w$W rc=$?"
done
# scenario suite on fast cores (fake 8-GPU cluster: policy/churn behavior)
timeout 600 python -m benchmarks.scenarios --scenario all > gpurun_out/scenarios8.json 2> gpurun_out/scenarios8.log
echo "scenarios rc=$?"
