"""ray.autoscaler.sdk parity: programmatic resource requests.

Role parity: reference python/ray/autoscaler/sdk.py request_resources —
a standing demand the autoscaler provisions for even before any task or
actor asks (stored in the GCS KV; StandardAutoscaler.pending_demands
merges it with the live pending actor/PG demands)."""
from __future__ import annotations

import json
from typing import Dict, List, Optional

_KV_NS = "autoscaler"
_KV_KEY = b"resource_request"


def request_resources(num_cpus: Optional[int] = None,
                      bundles: Optional[List[Dict[str, float]]] = None):
    """Set the cluster-wide standing resource request (replaces any
    previous request; pass nothing to clear it)."""
    from ant_ray_amd.experimental import internal_kv

    req: List[Dict[str, float]] = []
    if num_cpus:
        req.extend({"CPU": 1.0} for _ in range(int(num_cpus)))
    if bundles:
        req.extend(dict(b) for b in bundles)
    if req:
        internal_kv._internal_kv_put(
            _KV_KEY, json.dumps(req).encode(), namespace=_KV_NS.encode())
    else:
        internal_kv._internal_kv_del(_KV_KEY, namespace=_KV_NS.encode())


def get_requested_resources() -> List[Dict[str, float]]:
    from ant_ray_amd.experimental import internal_kv

    raw = internal_kv._internal_kv_get(_KV_KEY, namespace=_KV_NS.encode())
    if not raw:
        return []
    try:
        return json.loads(raw.decode())
    except Exception:
        return []
