#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 500 python bench.py --steps 200 --warmup 20 > gpurun_out/bench1.json 2> gpurun_out/bench1.log
echo "bench rc=$?"
timeout 300 python bench.py --steps 100 --warmup 10 --payload-every 1 > gpurun_out/bench_payload.json 2> gpurun_out/bench_payload.log
echo "bench-payload rc=$?"
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof2 -o membw --output-format csv -- \
  /root/repo/instaslice_amd/bin/instaslice-payload membw 2147483648 25 > /root/repo/gpurun_out/membw2.log 2>&1
tail -3 /root/repo/gpurun_out/membw2.log
