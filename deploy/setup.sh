#!/bin/bash
# Bootstrap a ROCm Kubernetes cluster for InstaSlice-AMD.
# Reference analog: deploy/setup.sh (KinD + NVIDIA GPU operator). The AMD
# path needs: the amdgpu driver on hosts, the AMD device plugin, and node
# labels; no CDI runtime-class tricks are required.
set -euo pipefail

# 1. AMD k8s device plugin (advertises amd.com/gpu; re-advertises partitions
#    after mode changes because KFD re-enumerates)
kubectl apply -f https://raw.githubusercontent.com/ROCm/k8s-device-plugin/master/k8s-ds-amdgpu-dp.yaml
kubectl apply -f deploy/device-plugin-configmap.yaml

# 2. node labeller (adds amd.com/gpu.* labels used by our nodeSelector)
kubectl apply -f https://raw.githubusercontent.com/ROCm/k8s-device-plugin/master/k8s-ds-amdgpu-labeller.yaml

# 3. our CRD + RBAC + controller + daemonset
kubectl apply -k config/default

# 4. mark GPU nodes
kubectl label nodes -l amd.com/gpu.family.AI amd.com/gpu.present=true --overwrite
echo "InstaSlice-AMD deployed. Try: kubectl apply -f samples/test-pod.yaml"
