#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
# bench with 4 controller workers (final r01 headline)
timeout 500 python bench.py --steps 300 --warmup 30 > gpurun_out/bench3.json 2> gpurun_out/bench3.log
echo "bench rc=$?"
# PMC: single counter first; then the pair if single works
export TMPDIR=/tmp
cd /tmp
timeout 200 rocprofv3 --pmc FETCH_SIZE --kernel-trace --stats -d /root/repo/gpurun_out/pmc1 -o fetch --output-format csv -- \
  /root/repo/instaslice_amd/bin/instaslice-payload membw 1073741824 5 0 1 > /root/repo/gpurun_out/pmc_fetch.log 2>&1
echo "pmc1 rc=$?"
timeout 200 rocprofv3 --pmc WRITE_SIZE --kernel-trace --stats -d /root/repo/gpurun_out/pmc2 -o write --output-format csv -- \
  /root/repo/instaslice_amd/bin/instaslice-payload membw 1073741824 5 0 1 > /root/repo/gpurun_out/pmc_write.log 2>&1
echo "pmc2 rc=$?"
