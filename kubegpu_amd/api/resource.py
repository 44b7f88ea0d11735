"""Request-name translation helpers (KubeDevice-API `resource` parity).

The reference calls resource.TranslateResource twice to wrap flat per-card
requests with the gpugrp0 / gpugrp1 grouping levels the node advertises
(gpuschedulerplugin/gpu.go:55-58).  The exact core implementation lives in
the external KubeDevice repo; we own the whole grammar here, so the
contract is defined in this module and honoured by scheduler synthesis
(kubegpu_amd.scheduler) and core binding (kubegpu_amd.core):

* Advertised node names are fully concrete:
      resource/group/gpugrp1/<H>/gpugrp0/<G>/gpu/<ID>/cards
* Request names may carry the wildcard index "*" at a group position:
      resource/group/gpugrp1/*/gpugrp0/*/gpu/0/cards
  meaning "any group at this level"; the binder resolves wildcards.
"""

from __future__ import annotations

import functools
import re
from typing import Tuple

from .types import ResourceList

WILDCARD = "*"


def node_advertises_level(node_resources: ResourceList, group_name: str) -> bool:
    """True when any advertised resource name contains /<group_name>/."""
    needle = f"/{group_name}/"
    return any(needle in name for name in node_resources)


def translate_resource(
    node_resources: ResourceList,
    container_requests: ResourceList,
    group_name: str,
    sub_name: str,
) -> bool:
    """Insert one grouping level into request names, in place.

    For every request name containing ``/<sub_name>/`` (or starting with
    ``<sub_name>/``) that does not already carry ``/<group_name>/``,
    rewrite ``.../<sub_name>/...`` to ``.../<group_name>/*/<sub_name>/...``
    — but only when the node's advertised resources actually use
    *group_name* (a flat node keeps flat requests).

    Parity: resource.TranslateResource(nodeRes, contReqs, groupName,
    subName) at gpu.go:55-58.  Returns True if any rewrite happened.
    """
    if not node_advertises_level(node_resources, group_name):
        return False
    changed = False
    pat = re.compile(rf"(^|/){re.escape(sub_name)}/")
    for name in list(container_requests.keys()):
        if f"/{group_name}/" in name:
            continue
        m = pat.search(name)
        if not m:
            continue
        new_name = (
            name[: m.start()]
            + m.group(1)
            + f"{group_name}/{WILDCARD}/{sub_name}/"
            + name[m.end():]
        )
        container_requests[new_name] = container_requests.pop(name)
        changed = True
    return changed


# Fully-concrete advertised leaf:  .../gpugrp1/H/gpugrp0/G/gpu/ID/cards
CARDS_RE = re.compile(r"^(?P<prefix>.*)/gpugrp1/(?P<h>[^/]+)/gpugrp0/(?P<g>[^/]+)/gpu/(?P<id>[^/]+)/cards$")


@functools.lru_cache(maxsize=8192)
def parse_cards_name(name: str) -> Tuple[str, str, str, str]:
    """Split an advertised/requested cards name into (prefix, h, g, id).

    Raises ValueError when the name is not a 2-level cards name.
    Memoized: the schedule hot path re-parses the same synthesized and
    advertised names constantly (names are drawn from a small set per
    cluster, so an LRU holds the working set).
    """
    m = CARDS_RE.match(name)
    if not m:
        raise ValueError(f"not a 2-level cards resource name: {name}")
    return m.group("prefix"), m.group("h"), m.group("g"), m.group("id")


def matches(request_name: str, concrete_name: str) -> bool:
    """Wildcard-aware match of a request name against a concrete name."""
    rp = request_name.split("/")
    cp = concrete_name.split("/")
    if len(rp) != len(cp):
        return False
    return all(r == WILDCARD or r == c for r, c in zip(rp, cp))
