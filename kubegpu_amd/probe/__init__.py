"""Bandwidth probes: RCCL-over-xGMI all-reduce + CDNA4 HBM kernels."""

from .bandwidth import (  # noqa: F401
    copy,
    d2d_copy_bw_gbps,
    load_ext,
    read_bw_gbps,
    write_bw_gbps,
)
from .rccl_probe import run_rccl_probe, torch_allreduce_busbw  # noqa: F401
from .xgmi_counters import (  # noqa: F401
    diff_link_metrics,
    probe_with_link_utilization,
    read_link_metrics,
)
