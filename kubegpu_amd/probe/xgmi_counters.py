"""xGMI link-utilization counters ("xGMI link utilization
during probe").

Wraps `amdsmiinfo linkmetrics` (csrc/amdsmiinfo.cpp over
amdsmi_get_link_metrics): per-link cumulative read/write KB.  Snapshot
before and after a probe run and diff to get bytes moved per physical
xGMI link — the direct check that traffic ran over the links the
scheduler's placement assumed.
"""

from __future__ import annotations

import json
import os
import subprocess
from typing import Dict, List, Optional


def _binary() -> str:
    env = os.environ.get("KUBEGPU_AMDSMIINFO")
    if env:
        return env
    here = os.path.dirname(os.path.abspath(__file__))
    return os.path.normpath(os.path.join(here, "..", "csrc", "bin", "amdsmiinfo"))


def read_link_metrics(binary: Optional[str] = None, timeout_s: float = 30.0) -> Dict:
    """One snapshot: {gpu index: [{link, type, read_kb, write_kb}, ...]}."""
    out = subprocess.run(
        [binary or _binary(), "linkmetrics"],
        capture_output=True,
        timeout=timeout_s,
        check=True,
    )
    raw = json.loads(out.stdout.decode())
    return {g["index"]: g["links"] for g in raw["gpus"]}


def diff_link_metrics(before: Dict, after: Dict) -> Dict[int, List[Dict]]:
    """Per-link traffic between two snapshots (MB moved per link)."""
    result: Dict[int, List[Dict]] = {}
    for idx, links_after in after.items():
        links_before = {l["link"]: l for l in before.get(idx, [])}
        out = []
        for l in links_after:
            b = links_before.get(l["link"], {"read_kb": 0, "write_kb": 0})
            out.append(
                {
                    "link": l["link"],
                    "type": l.get("type"),
                    "read_mb": (l["read_kb"] - b["read_kb"]) / 1024.0,
                    "write_mb": (l["write_kb"] - b["write_kb"]) / 1024.0,
                }
            )
        result[idx] = out
    return result


def probe_with_link_utilization(probe_fn, *args, **kwargs):
    """Run *probe_fn* bracketed by counter snapshots.

    Returns (probe result, per-link traffic diff).  Falls back to
    (result, None) when counters are unavailable (e.g. no amdsmi).
    """
    try:
        before = read_link_metrics()
    except Exception:
        before = None
    result = probe_fn(*args, **kwargs)
    if before is None:
        return result, None
    try:
        after = read_link_metrics()
    except Exception:
        return result, None
    return result, diff_link_metrics(before, after)
