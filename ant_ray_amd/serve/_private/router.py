"""Router + DeploymentHandle plumbing.

Role parity: reference python/ray/serve/_private/router.py:496
(AsyncioRouter) and request_router/pow_2_router.py:52 — power-of-two-choices
replica selection on queue length. The replica list is pulled from the
controller and cached (refreshed periodically or when every replica looks
dead), instead of the reference's long-poll push.
"""
from __future__ import annotations

import random
import time
from typing import Any, Dict, List, Optional

from ant_ray_amd.serve._private.common import CONTROLLER_NAME

_REFRESH_S = 5.0


class Router:
    def __init__(self, app: str, deployment: str):
        self.app = app
        self.deployment = deployment
        self._replicas: List[Any] = []
        self._last_refresh = 0.0
        self._ongoing: Dict[int, int] = {}  # index -> in-flight from THIS router

    def _controller(self):
        import ant_ray_amd as ray

        return ray.get_actor(CONTROLLER_NAME)

    def _refresh(self, force=False):
        now = time.monotonic()
        if not force and self._replicas and now - self._last_refresh < _REFRESH_S:
            return
        import ant_ray_amd as ray

        replicas = ray.get(
            self._controller().get_replicas.remote(self.app, self.deployment)
        )
        if replicas is None:
            raise RuntimeError(
                f"deployment '{self.deployment}' not found in app '{self.app}'"
            )
        try:
            info = ray.get(self._controller().get_deployment_info.remote(
                self.app, self.deployment))
            self._max_ongoing = int(
                (info or {}).get("max_ongoing_requests") or 0) or None
        except Exception:
            self._max_ongoing = None
        self._replicas = replicas
        self._ongoing = {i: self._ongoing.get(i, 0) for i in range(len(replicas))}
        self._last_refresh = now

    def _admit(self, i: int, deadline_s: float = 60.0) -> int:
        """Per-replica max_ongoing_requests backpressure (reference
        router honors the deployment cap): when the chosen replica is at
        capacity, spill to the least-loaded one; when EVERY replica is
        at capacity, wait until one drains."""
        cap = getattr(self, "_max_ongoing", None)
        if not cap:
            return i
        deadline = time.monotonic() + deadline_s
        while True:
            if self._ongoing.get(i, 0) < cap:
                return i
            j = min(range(len(self._replicas)),
                    key=lambda k: self._ongoing.get(k, 0))
            if self._ongoing.get(j, 0) < cap:
                return j
            if time.monotonic() > deadline:
                raise RuntimeError(
                    f"all {len(self._replicas)} replicas of "
                    f"{self.app}/{self.deployment} at max_ongoing_requests="
                    f"{cap} for {deadline_s:.0f}s")
            time.sleep(0.005)

    def choose_replica(self):
        """Power-of-two-choices on locally tracked in-flight counts."""
        self._refresh()
        n = len(self._replicas)
        if n == 0:
            self._refresh(force=True)
            n = len(self._replicas)
            if n == 0:
                raise RuntimeError(
                    f"no replicas for {self.app}/{self.deployment}")
        if n == 1:
            i = 0
        else:
            a, b = random.sample(range(n), 2)
            i = a if self._ongoing.get(a, 0) <= self._ongoing.get(b, 0) else b
        return i, self._replicas[i]

    def submit(self, method_name: Optional[str], args, kwargs,
               multiplexed_model_id: Optional[str] = None):
        """Returns (ObjectRef, replica_index). The DeploymentResponse calls
        complete(index) when the result is consumed, closing the in-flight
        accounting the pow-2 choice reads. multiplexed_model_id biases
        routing to the replica that served that model last (parity:
        the reference's model-multiplex-aware router — avoids reloading
        an LRU-cached model on a cold replica)."""
        i = None
        if multiplexed_model_id is not None:
            cache = getattr(self, "_model_affinity", None)
            if cache is None:
                cache = self._model_affinity = {}
            self._refresh()
            j = cache.get(multiplexed_model_id)
            if j is not None and j < len(self._replicas):
                i = j
        if i is None:
            i, _ = self.choose_replica()
        i = self._admit(i)
        if multiplexed_model_id is not None:
            self._model_affinity[multiplexed_model_id] = i
            while len(self._model_affinity) > 1024:
                self._model_affinity.pop(next(iter(self._model_affinity)))
        replica = self._replicas[i]
        self._ongoing[i] = self._ongoing.get(i, 0) + 1
        return replica.handle_request.remote(method_name, args, kwargs), i

    def submit_stream(self, method_name: Optional[str], args, kwargs):
        """Streaming variant: returns (ObjectRefGenerator, replica_index)."""
        i, replica = self.choose_replica()
        i = self._admit(i)
        replica = self._replicas[i]
        self._ongoing[i] = self._ongoing.get(i, 0) + 1
        return replica.handle_request_streaming.remote(
            method_name, args, kwargs), i

    def complete(self, i: int):
        self._ongoing[i] = max(0, self._ongoing.get(i, 1) - 1)
