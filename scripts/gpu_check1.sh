#!/bin/bash
# First GPU-box validation: environment facts, gpu pytest tier, 1-GPU bench,
# rocprof stats on the payload kernels.
set -x
cd /root/repo
mkdir -p gpurun_out
{
  rocm-smi --showproductname 2>&1 | head -20
  amd-smi list 2>&1 | head -40
  amd-smi partition 2>&1 | head -60
} > gpurun_out/env_facts.txt 2>&1

python - <<'PY' > gpurun_out/enum.json 2> gpurun_out/enum.err
import json
from instaslice_amd.smi import _partitiond as pd
d = pd.Partitiond(); d.init()
procs = d.enumerate()
out = [dict(index=p.index, uuid=p.uuid, asic_name=p.asic_name,
            asic_serial=p.asic_serial, vram_total_mb=p.vram_total_mb,
            bdf=p.bdf, node_id=p.node_id, partition_id=p.partition_id,
            compute=p.compute_partition, memory=p.memory_partition,
            cus=p.num_compute_units) for p in procs]
try:
    profs = d.get_profile_config(0)
    out.append({"profiles": [dict(t=x.profile_type, n=x.num_partitions,
               idx=x.profile_index, caps=list(x.memory_caps)) for x in profs]})
except Exception as e:
    out.append({"profiles_error": str(e)})
print(json.dumps(out, indent=1))
PY

timeout 600 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/pytest_gpu.log

timeout 400 python bench.py --steps 60 --warmup 10 > gpurun_out/bench1.json 2> gpurun_out/bench1.log

export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o payload -- \
  /root/repo/instaslice_amd/bin/instaslice-payload membw 1073741824 20 \
  > /root/repo/gpurun_out/membw_prof.log 2>&1
echo "=== DONE rc=$? ==="
