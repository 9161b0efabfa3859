"""InstaSlice-AMD: MI355X-native dynamic GPU-partition operator.

A brand-new framework with the capabilities of project-codeflare/instaslice
(reference: /root/reference), re-designed for AMD Instinct MI355X:

- Partitioning is *mode-based and whole-GPU* (SPX/DPX/QPX/CPX compute modes x
  NPS1/NPS4 memory modes set via amd-smi), not slot-based MIG carving. The
  placer therefore plans per-GPU mode transitions instead of per-slot
  placement (reference: internal/controller/instaslice_controller.go:303-384
  packs {start,size} slots; here the analog is an XCD-ordinal bitmap under a
  per-GPU mode constraint).
- The native device layer is first-party C++ against /opt/rocm/lib/libamd_smi.so
  ("partitiond"), replacing the reference's go-nvml cgo bindings
  (instaslice_daemonset.go:29, go.mod:22). Enumeration is cached once, fixing
  the reference's per-reconcile nvml.Init (instaslice_daemonset.go:112).
- Workload validation payloads are hand-written gfx950 HIP kernels
  (instaslice_amd/ops/csrc/payload.hip), the analog of the reference's
  cuda-vectoradd sample container (samples/test-pod.yaml:12).

Layer map (mirrors SURVEY.md section 1):
  L0 api/        CRD-shaped data model (Instaslice kind, group
                 inference.codeflare.dev/v1alpha1 kept for API compatibility)
  L1 controller/ cluster controller: gated-pod admission, placement, ungating
  L2 agent/      per-node daemonset: realizes partitions via amd-smi
  L3 smi/        device layer: C++ partitiond bindings + FakeAmdSmi test double
  L4 config/, deploy/, samples/  packaging + k8s manifests (repo root)

The coordination bus is a CR-shaped state store (store/) with watch semantics:
in-memory for tests, TCP-served for multi-rank benchmarks, and the same
interface a real API-server adapter implements in a cluster.
"""

__version__ = "0.1.0"

API_GROUP = "inference.codeflare.dev"
API_VERSION = "v1alpha1"

# Pod-contract constants, kept byte-compatible with the reference so existing
# workload YAMLs keep working (reference: samples/test-pod.yaml:1-21;
# gate/finalizer name incl. original spelling at instaslice_controller.go:386-395).
GATE_NAME = "org.instaslice/accelarator"
FINALIZER_NAME = "org.instaslice/accelarator"
RESOURCE_PREFIX = "amd.com/"  # e.g. limits: {"amd.com/cpx-1x36": 1}
POD_RESOURCE_PREFIX = "org.instaslice/"  # per-pod extended resource pinning
# set on a gated pod when no node can currently fit its profile (the
# reference requeues silently, instaslice_controller.go:231; surfacing the
# condition lets schedulers/users react instead of waiting blind)
UNSCHEDULABLE_ANNOTATION = "org.instaslice/unschedulable"
PRIORITY_ANNOTATION = "org.instaslice/priority"  # int; higher may preempt
GROUP_SIZE_ANNOTATION = "org.instaslice/group-size"  # gang ungate barrier
