"""ray.experimental.internal_kv parity: direct GCS KV access.

Role parity: reference python/ray/experimental/internal_kv.py (used by
libraries for cluster-global metadata; backed by gcs_kv_manager.cc)."""
from __future__ import annotations

from typing import List, Optional


def _cw():
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    if cw is None or not cw.connected:
        raise RuntimeError("internal_kv requires ray.init()")
    return cw


def _internal_kv_initialized() -> bool:
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    return cw is not None and cw.connected


def _internal_kv_put(key: bytes, value: bytes, overwrite: bool = True,
                     namespace: Optional[bytes] = None) -> bool:
    """Returns True if the key was already present."""
    cw = _cw()
    ns = (namespace or b"").decode() if isinstance(namespace, bytes) else (namespace or "")
    r = cw.io.run(cw.gcs.call("kv_put", {
        "ns": ns, "key": bytes(key), "value": bytes(value),
        "overwrite": overwrite}))
    return not r.get("added", False)


def _internal_kv_get(key: bytes, namespace: Optional[bytes] = None) -> Optional[bytes]:
    cw = _cw()
    ns = (namespace or b"").decode() if isinstance(namespace, bytes) else (namespace or "")
    r = cw.io.run(cw.gcs.call("kv_get", {"ns": ns, "key": bytes(key)}))
    return r.get("value")


def _internal_kv_del(key: bytes, namespace: Optional[bytes] = None) -> bool:
    cw = _cw()
    ns = (namespace or b"").decode() if isinstance(namespace, bytes) else (namespace or "")
    r = cw.io.run(cw.gcs.call("kv_del", {"ns": ns, "key": bytes(key)}))
    return r.get("deleted", False)


def _internal_kv_list(prefix: bytes, namespace: Optional[bytes] = None) -> List[bytes]:
    cw = _cw()
    ns = (namespace or b"").decode() if isinstance(namespace, bytes) else (namespace or "")
    r = cw.io.run(cw.gcs.call("kv_keys", {"ns": ns, "prefix": bytes(prefix)}))
    return r.get("keys", [])


def _internal_kv_exists(key: bytes, namespace: Optional[bytes] = None) -> bool:
    return _internal_kv_get(key, namespace) is not None
