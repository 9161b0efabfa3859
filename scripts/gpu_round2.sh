#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/pytest_gpu2.log

timeout 500 python bench.py --steps 300 --warmup 30 > gpurun_out/bench2.json 2> gpurun_out/bench2.log
echo "bench rc=$?"

# membw block sweep (2 GiB buffers)
for B in 1024 2048 4096 8192 16384 32768 65535; do
  timeout 120 ./instaslice_amd/bin/instaslice-payload membw 2147483648 20 $B
done > gpurun_out/membw_sweep.json 2>&1

# PMC counter evidence for the copy kernel (counters + stats only)
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --kernel-trace --stats \
  -d /root/repo/gpurun_out/pmc -o membw_pmc --output-format csv -- \
  /root/repo/instaslice_amd/bin/instaslice-payload membw 2147483648 10 \
  > /root/repo/gpurun_out/membw_pmc.log 2>&1
echo "pmc rc=$?"
