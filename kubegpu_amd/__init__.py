"""kubegpu_amd — MI355X-native Kubernetes GPU device plugin + topology-aware scheduler.

A from-scratch AMD-native re-design of the capabilities of microsoft/KubeGPU
(reference surveyed in SURVEY.md): GPU discovery via amdsmi (gfx950 device
IDs, 288 GB HBM3E, per-link xGMI graph), a topology scheduler that ranks
candidate GPU subsets by xGMI ring bandwidth instead of interconnect-level
trees, and container allocation that injects /dev/kfd + /dev/dri render
nodes with ROCR_VISIBLE_DEVICES.

Subpackages
-----------
api             re-owned KubeDevice-API surface (types, Device,
                DeviceScheduler interfaces; cf. reference imports at
                gpuschedulerplugin/gpu.go:8-10)
plugintypes     shared tree types (cf. gpuplugintypes/)
discovery       amdsmi / sysfs / fake GPU enumeration backends
                (cf. nvidiagpuplugin/gpu/nvml/, nvgputypes/)
deviceplugin    node-side device manager (cf. nvidiagpuplugin/gpu/nvidia/)
scheduler       topology-aware device scheduler (cf. gpuschedulerplugin/)
core            minimal group-scheduler core — the half the reference left
                to the external KubeDevice repo; we own both sides
probe           RCCL-over-xGMI bandwidth probe + HIP HBM bandwidth kernels
cli             operator CLIs (cf. nvidiagpuplugin/cmd/, nvmlinfo/)
"""

__version__ = "0.2.0"
