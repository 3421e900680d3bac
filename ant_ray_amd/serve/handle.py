"""DeploymentHandle / DeploymentResponse.

Role parity: reference python/ray/serve/handle.py:757 (DeploymentHandle,
.remote :833 → router) and DeploymentResponse (awaitable, .result()).
Handles are picklable by (app, deployment) and rebuild their router lazily
in the borrowing process (replica actor or driver).
"""
from __future__ import annotations

from typing import Any, Optional

from ant_ray_amd.serve._private.router import Router


class DeploymentResponse:
    def __init__(self, ref, router: Optional[Router] = None,
                 replica_index: Optional[int] = None):
        self._ref = ref
        self._router = router
        self._index = replica_index
        self._completed = False

    def _complete(self):
        if not self._completed and self._router is not None:
            self._completed = True
            self._router.complete(self._index)

    def result(self, timeout_s: Optional[float] = None) -> Any:
        import ant_ray_amd as ray

        try:
            return ray.get(self._ref, timeout=timeout_s)
        finally:
            self._complete()

    def __await__(self):
        def gen():
            try:
                out = yield from self._ref.__await__()
                return out
            finally:
                self._complete()

        return gen()

    def _to_object_ref(self):
        self._complete()
        return self._ref

    def __del__(self):
        try:
            self._complete()
        except Exception:
            pass


class DeploymentResponseGenerator:
    """Iterator over a streaming deployment call's chunk values (parity:
    reference DeploymentResponseGenerator from handle.options(stream=True);
    here iteration yields the chunk VALUES directly)."""

    def __init__(self, gen, router, index):
        self._gen = gen
        self._router = router
        self._index = index

    def __iter__(self):
        return self

    def __next__(self):
        import ant_ray_amd as ray

        try:
            return ray.get(next(self._gen))
        except StopIteration:
            self._router.complete(self._index)
            raise

    def __aiter__(self):
        return self

    async def __anext__(self):
        import ant_ray_amd as ray

        try:
            ref = await self._gen.__anext__()
        except StopAsyncIteration:
            self._router.complete(self._index)
            raise
        return ray.get(ref)


class DeploymentHandle:
    def __init__(self, deployment_name: str, app_name: str = "default",
                 method_name: Optional[str] = None):
        self.deployment_name = deployment_name
        self.app_name = app_name
        self._method_name = method_name
        self._router: Optional[Router] = None

    def _get_router(self) -> Router:
        if self._router is None:
            self._router = Router(self.app_name, self.deployment_name)
        return self._router

    def options(self, *, method_name: Optional[str] = None,
                stream: bool = False,
                multiplexed_model_id: Optional[str] = None, **_):
        h = DeploymentHandle(self.deployment_name, self.app_name,
                             method_name or self._method_name)
        h._stream = stream
        h._model_id = multiplexed_model_id
        # share ONE router across derived handles: in-flight accounting
        # and model-affinity state must not reset per .options() call
        h._router = self._get_router()
        return h

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return DeploymentHandle(self.deployment_name, self.app_name, name)

    def remote(self, *args, **kwargs):
        router = self._get_router()
        if getattr(self, "_stream", False):
            gen, idx = router.submit_stream(self._method_name, args, kwargs)
            return DeploymentResponseGenerator(gen, router, idx)
        ref, i = router.submit(
            self._method_name, args, kwargs,
            multiplexed_model_id=getattr(self, "_model_id", None))
        return DeploymentResponse(ref, router, i)

    def __reduce__(self):
        return (DeploymentHandle,
                (self.deployment_name, self.app_name, self._method_name))

    def __repr__(self):
        return (f"DeploymentHandle(app={self.app_name!r}, "
                f"deployment={self.deployment_name!r})")
