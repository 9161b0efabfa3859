#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out

# NT vs regular copy sweep
for NT in 0 1; do
  for B in 1024 4096 16384 65535; do
    timeout 120 ./instaslice_amd/bin/instaslice-payload membw 2147483648 20 $B $NT
  done
done > gpurun_out/membw_nt_sweep.json 2>&1

# what counters exist on gfx950?
export TMPDIR=/tmp
cd /tmp
timeout 120 rocprofv3 --list-avail > /root/repo/gpurun_out/counters_avail.txt 2>&1
grep -iE 'FETCH|WRITE|TCC|SQ_BUSY|GRBM' /root/repo/gpurun_out/counters_avail.txt | head -40 > /root/repo/gpurun_out/counters_mem.txt
