"""CDI (Container Device Interface) spec generation.

The reference-era injection path was a vendor runtime hook
(nvidia-docker); this build's device-plugin path injects /dev nodes
directly.  Modern container runtimes (containerd >= 1.7, CRI-O) also
accept a declarative CDI spec, which lets ANY runtime consumer — not
just our kubelet plugin — request `amd.com/gpu=<uuid>` and get the
right device nodes with no hook.  This module renders that spec from
discovery:

* per-GPU device entries: the GPU's /dev/dri/renderD* (and card) node;
* a common edit adding /dev/kfd (the compute interface every ROCm
  process needs) and the ROCR_VISIBLE_DEVICES env for the chosen GPU.

`amddevs --cdi` prints the spec; write it to
/etc/cdi/amd.com-gpu.json (or .yaml) on the node to activate.
Spec format: CDI v0.6.0 (github.com/cncf-tags/container-device-interface).
"""

from __future__ import annotations

import json
from typing import Dict, List

from ..discovery import GpusInfo

CDI_VERSION = "0.6.0"
CDI_KIND = "amd.com/gpu"


def _device_nodes(paths: List[str]) -> List[Dict]:
    return [{"path": p, "hostPath": p, "permissions": "rw"} for p in paths if p]


def cdi_spec(info: GpusInfo) -> Dict:
    """Render the CDI spec dict for a node's discovered GPUs."""
    devices = []
    for g in sorted(info.devices, key=lambda d: d.index):
        edits: Dict = {
            "deviceNodes": _device_nodes([g.render_path, g.card_path]),
            "env": [f"ROCR_VISIBLE_DEVICES={g.uuid}"],
        }
        devices.append({"name": g.uuid, "containerEdits": edits})
        # index alias: `amd.com/gpu=0` works like the uuid form
        devices.append(
            {"name": str(g.index), "containerEdits": json.loads(json.dumps(edits))}
        )
    spec = {
        "cdiVersion": CDI_VERSION,
        "kind": CDI_KIND,
        # /dev/kfd is shared by every GPU: one common edit, not per-device
        "containerEdits": {"deviceNodes": _device_nodes(["/dev/kfd"])},
        "devices": devices,
    }
    return spec


def cdi_json(info: GpusInfo) -> str:
    return json.dumps(cdi_spec(info), indent=1)
