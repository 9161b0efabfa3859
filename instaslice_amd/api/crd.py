"""CRD manifest generation: the controller-gen analog.

Emits the OpenAPI v3 schema for the Instaslice kind from the same type
definitions the code uses (api/types.py), so config/crd/ can never drift
from the implementation (the reference generates this with controller-gen
from Go markers; config/crd/bases/inference.codeflare.dev_instaslices.yaml).
tests/test_manifests.py asserts the checked-in YAML equals this output.
"""

from __future__ import annotations

from instaslice_amd import API_GROUP, API_VERSION

_STR = {"type": "string"}
_INT = {"type": "integer"}
_BOOL = {"type": "boolean"}

ALLOCATION_SCHEMA = {
    "type": "object",
    "required": ["profile", "gpuUUID", "ordinal", "start", "size", "podUUID",
                 "podName", "namespace", "nodename"],
    "properties": {
        "profile": _STR,
        "gpuUUID": _STR,
        "ordinal": _INT,
        "start": _INT,
        "size": _INT,
        "podUUID": _STR,
        "podName": _STR,
        "namespace": _STR,
        "nodename": _STR,
        "allocationStatus": {
            "type": "string",
            "enum": ["creating", "created", "ungated", "deleted", "failed"],
        },
        "computeMode": _STR,
        "memoryMode": _STR,
        "group": _STR,
        "priority": _INT,
    },
}

PREPARED_SCHEMA = {
    "type": "object",
    "required": ["parentGpuUUID", "ordinal", "computeMode", "memoryMode",
                 "xcds", "memoryGB", "podUUID"],
    "properties": {
        "parentGpuUUID": _STR,
        "ordinal": _INT,
        "computeMode": _STR,
        "memoryMode": _STR,
        "xcds": _INT,
        "memoryGB": _INT,
        "podUUID": _STR,
        "deviceIndex": _INT,
    },
}

GPU_SCHEMA = {
    "type": "object",
    "properties": {
        "uuid": _STR,
        "model": _STR,
        "memoryGB": _INT,
        "computeMode": {"type": "string",
                        "enum": ["SPX", "DPX", "TPX", "QPX", "CPX"]},
        "memoryMode": {"type": "string",
                       "enum": ["NPS1", "NPS2", "NPS4"]},
        "usedOrdinals": {"type": "array", "items": _INT},
        "index": _INT,
        "modeLocked": _BOOL,
        # drain-time mode hint (controller writes, agent pre-flips idle GPUs)
        "desiredMode": {"type": "string",
                        "enum": ["SPX", "DPX", "TPX", "QPX", "CPX"]},
    },
}

PROFILE_SCHEMA = {
    "type": "object",
    "properties": {
        "name": _STR,
        "compute": _STR,
        "xcds": _INT,
        "memory_gb": _INT,
        "preferred_memory": _STR,
        "profile_index": {"type": "integer", "nullable": True},
    },
}

PLACEMENTS_SCHEMA = {
    "type": "object",
    "properties": {
        "gpu_model": _STR,
        "total_memory_gb": _INT,
        "xcd_count": _INT,
        "profiles": {"type": "array", "items": PROFILE_SCHEMA},
    },
}

SPEC_SCHEMA = {
    "type": "object",
    "properties": {
        "gpuUuids": {"type": "object", "additionalProperties": _STR},
        "gpus": {"type": "object", "additionalProperties": GPU_SCHEMA},
        "placements": PLACEMENTS_SCHEMA,
        "allocations": {"type": "object",
                        "additionalProperties": ALLOCATION_SCHEMA},
        "prepared": {"type": "object",
                     "additionalProperties": PREPARED_SCHEMA},
        # preemption slot reservations (controller/reconciler._maybe_preempt)
        "nominations": {
            "type": "object",
            "additionalProperties": {
                "type": "object",
                "properties": {
                    "gpuUUID": _STR,
                    "ordinal": _INT,
                    "ts": {"type": "number"},
                    # whole-GPU nomination (plan 2 preemption): the GPU is
                    # reserved for a mode-flip placement once it drains
                    "wholeGpu": _BOOL,
                },
            },
        },
        # agent opts into two-phase teardown when it must reset the GPU
        # mode itself (agent/daemonset.py writes this at boot)
        "agentManagedTeardown": _BOOL,
        # xGMI link topology discovered at boot: {src_uuid: {dst_uuid: hops}}
        # (amdsmi_topo_get_link_type); gang scoring prefers 1-hop neighbors
        "topology": {
            "type": "object",
            "additionalProperties": {
                "type": "object",
                "additionalProperties": _INT,
            },
        },
        # operator drain (CLI cordon): no new placements on this node
        "cordoned": _BOOL,
    },
}


def crd_manifest() -> dict:
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"instaslices.{API_GROUP}"},
        "spec": {
            "group": API_GROUP,
            "names": {
                "kind": "Instaslice",
                "listKind": "InstasliceList",
                "plural": "instaslices",
                "singular": "instaslice",
            },
            "scope": "Namespaced",
            "versions": [
                {
                    "name": API_VERSION,
                    "served": True,
                    "storage": True,
                    "subresources": {"status": {}},
                    "schema": {
                        "openAPIV3Schema": {
                            "type": "object",
                            "properties": {
                                "apiVersion": _STR,
                                "kind": _STR,
                                "metadata": {"type": "object"},
                                "spec": SPEC_SCHEMA,
                                "status": {
                                    "type": "object",
                                    "properties": {
                                        "processed": _STR,
                                        "heartbeat": {"type": "number"},
                                        "gpuMetrics": {
                                            "type": "object",
                                            "additionalProperties": {
                                                "type": "object",
                                                "additionalProperties": {
                                                    "type": "number"},
                                            },
                                        },
                                    },
                                },
                            },
                        }
                    },
                }
            ],
        },
    }


def crd_yaml() -> str:
    import yaml

    return (
        "# Generated by instaslice_amd.api.crd — do not edit by hand.\n"
        "# Regenerate: python -m instaslice_amd.api.crd > "
        "config/crd/bases/inference.codeflare.dev_instaslices.yaml\n"
        + yaml.safe_dump(crd_manifest(), sort_keys=False)
    )


if __name__ == "__main__":
    print(crd_yaml(), end="")
