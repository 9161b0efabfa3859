#!/bin/bash
# README-runbook transcript on real MI355X hardware: the reference's manual
# verification flow (README.md:188-300) as a scripted demo. Output lands in
# gpurun_out/runbook.txt and is committed under profiles/.
set -x
cd /root/repo
mkdir -p gpurun_out
OUT=gpurun_out/runbook.txt
: > $OUT

log() { echo -e "\n\$ $*" >> $OUT; }

log "python build_native.py"
timeout 300 python build_native.py >> $OUT 2>&1

log "python -m instaslice_amd controlplane --port 7080 --grace 0 &"
timeout 280 python -m instaslice_amd controlplane --port 7080 --grace 0 >> $OUT 2>&1 &
CP=$!
sleep 3

log "python -m instaslice_amd daemonset --store 127.0.0.1:7080 --node-name mi355x-node --metrics-port 18084 --devplugin &"
timeout 270 python -m instaslice_amd daemonset --store 127.0.0.1:7080 \
  --node-name mi355x-node --metrics-port 18084 --devplugin >> $OUT 2>&1 &
DS=$!
sleep 5

log "python -m instaslice_amd status"
timeout 30 python -m instaslice_amd status --store 127.0.0.1:7080 >> $OUT 2>&1

# profile matching the GPU's current mode (flips are platform-refused here)
PROFILE=$(timeout 30 python - <<'EOF'
from instaslice_amd.store.netstore import NetStoreClient
from instaslice_amd.partition.profiles import ProfileCatalog
c = NetStoreClient("127.0.0.1", 7080)
cr = c.get("Instaslice", "mi355x-node", "instaslice-system")
cat = ProfileCatalog.from_dict(cr["spec"]["placements"])
mode = next(iter(cr["spec"]["gpus"].values()))["computeMode"]
print(next(p.name for p in cat.profiles if p.compute.value == mode))
c.close()
EOF
)
echo "chosen profile: $PROFILE" >> $OUT

log "python -m instaslice_amd submit --name demo --profile $PROFILE --wait"
timeout 60 python -m instaslice_amd submit --store 127.0.0.1:7080 \
  --name demo --profile $PROFILE --wait --timeout 30 >> $OUT 2>&1

log "python -m instaslice_amd status   (allocation realized)"
timeout 30 python -m instaslice_amd status --store 127.0.0.1:7080 >> $OUT 2>&1

log "run the HIP payload inside the pod's partition (vecadd + device info)"
DEV=$(timeout 30 python - <<'EOF'
from instaslice_amd.store.netstore import NetStoreClient
c = NetStoreClient("127.0.0.1", 7080)
cm = c.get("ConfigMap", "demo", "default")
print(cm["data"]["ROCR_VISIBLE_DEVICES"])
c.close()
EOF
)
ROCR_VISIBLE_DEVICES=$DEV timeout 60 ./instaslice_amd/bin/instaslice-payload vecadd 4194304 >> $OUT 2>&1
ROCR_VISIBLE_DEVICES=$DEV timeout 60 ./instaslice_amd/bin/instaslice-payload info >> $OUT 2>&1

log "device-plugin shim advertisement (amd.com/gpu on node capacity)"
timeout 30 python - <<'EOF2' >> $OUT 2>&1
from instaslice_amd.store.netstore import NetStoreClient
c = NetStoreClient("127.0.0.1", 7080)
node = c.get("Node", "mi355x-node", "")
print("node capacity:", node["status"].get("capacity"))
c.close()
EOF2

log "SERVING WORKLOAD inside the partition (bf16 transformer decode)"
ROCR_VISIBLE_DEVICES=$DEV HIP_VISIBLE_DEVICES=$DEV timeout 240 \
  python -m instaslice_amd.ops.serving_check --layers 4 --dmodel 1024 \
  --prefill 256 --decode 16 >> $OUT 2>&1

log "curl the agent metrics endpoint"
timeout 10 curl -s http://127.0.0.1:18084/metrics | grep -E "instaslice_(alloc|partition|reconcile)" | head -12 >> $OUT 2>&1

log "python -m instaslice_amd describe --name demo   (events timeline)"
timeout 30 python -m instaslice_amd describe --store 127.0.0.1:7080 --name demo >> $OUT 2>&1

log "python -m instaslice_amd delete --name demo"
timeout 30 python -m instaslice_amd delete --store 127.0.0.1:7080 --name demo >> $OUT 2>&1
sleep 2

log "PREEMPTION DEMO: fill the node at priority 1, then submit priority 10"
timeout 60 python - <<PYEOF >> $OUT 2>&1
import time
from instaslice_amd.store.netstore import NetStoreClient
from instaslice_amd.api.types import new_pod
c = NetStoreClient("127.0.0.1", 7080)
c.create(new_pod("low", profile="$PROFILE", priority=1))
deadline = time.monotonic() + 20
while time.monotonic() < deadline:
    if not c.get("Pod", "low", "default")["spec"].get("schedulingGates"):
        break
    time.sleep(0.05)
print("low-priority pod holds the partition")
t0 = time.perf_counter()
c.create(new_pod("high", profile="$PROFILE", priority=10))
while time.monotonic() < deadline:
    try:
        if not c.get("Pod", "high", "default")["spec"].get("schedulingGates"):
            break
    except Exception:
        pass
    time.sleep(0.01)
print("high-priority pod scheduled in %.1f ms (low evicted)"
      % ((time.perf_counter() - t0) * 1e3))
for name in ("high.Preempting", "low.Preempted"):
    ev = c.get("Event", name, "default")
    print(f"event {name}: {ev['message']}")
c.delete("Pod", "high", "default")
c.close()
PYEOF

log "CORDON DEMO: drain the node, submit, uncordon"
timeout 30 python -m instaslice_amd cordon --store 127.0.0.1:7080 --node mi355x-node >> $OUT 2>&1
timeout 30 python -m instaslice_amd submit --store 127.0.0.1:7080 --name held --profile $PROFILE >> $OUT 2>&1
sleep 1
timeout 30 python -m instaslice_amd describe --store 127.0.0.1:7080 --name held | grep -A3 Unschedulable >> $OUT 2>&1
timeout 30 python -m instaslice_amd cordon --store 127.0.0.1:7080 --node mi355x-node --uncordon >> $OUT 2>&1
timeout 60 python -m instaslice_amd submit --store 127.0.0.1:7080 --name held2 --profile $PROFILE --wait --timeout 30 > /dev/null 2>&1
echo "uncordoned; placements resumed" >> $OUT
timeout 30 python -m instaslice_amd delete --store 127.0.0.1:7080 --name held >> $OUT 2>&1
timeout 30 python -m instaslice_amd delete --store 127.0.0.1:7080 --name held2 >> $OUT 2>&1
sleep 1

log "python -m instaslice_amd status   (drained)"
timeout 30 python -m instaslice_amd status --store 127.0.0.1:7080 >> $OUT 2>&1

kill $DS $CP 2>/dev/null
wait 2>/dev/null
echo "runbook done rc=0"
