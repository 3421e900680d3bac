"""ServeController actor: app/deployment/replica state reconciliation.

Role parity: reference python/ray/serve/_private/controller.py:105
(ServeController singleton actor), application_state.py /
deployment_state.py (state machines), autoscaling_policy.py:16
(_calculate_desired_num_replicas). Config push is pull-based here (handles/
proxies refresh replica lists from the controller, parity long_poll.py in
spirit), replica fan-out and health checks are asyncio background loops in
this actor.
"""
from __future__ import annotations

import asyncio
import logging
import pickle
import time
from typing import Any, Dict, List, Optional

import ant_ray_amd as ray
from ant_ray_amd.serve._private.common import AutoscalingConfig
from ant_ray_amd.serve._private.replica import Replica

logger = logging.getLogger("antray.serve.controller")


def _calculate_desired_num_replicas(cfg: AutoscalingConfig,
                                    total_ongoing: float,
                                    current: int) -> int:
    """Parity: serve/autoscaling_policy.py:16 — replicas sized so each sees
    ~target_ongoing_requests."""
    if current == 0:
        return cfg.min_replicas
    desired = total_ongoing / max(cfg.target_ongoing_requests, 1e-9)
    import math

    desired = math.ceil(desired)
    return max(cfg.min_replicas, min(cfg.max_replicas, desired))


class DeploymentReplicas:
    def __init__(self):
        self.replicas: List[Any] = []  # actor handles
        self.target: int = 0
        self.cfg: Dict[str, Any] = {}
        self.last_scale_up = 0.0
        self.last_scale_down = 0.0
        self.ongoing_history: List[float] = []


class ServeController:
    """Detached named actor; all methods are async (max_concurrency high)."""

    def __init__(self, http_port: int = 8000, http_host: str = "127.0.0.1"):
        # {app: {deployment: DeploymentReplicas}}
        self.apps: Dict[str, Dict[str, DeploymentReplicas]] = {}
        self.ingress: Dict[str, str] = {}  # app -> ingress deployment name
        self.route_prefixes: Dict[str, str] = {}  # app -> route prefix
        self.http_port = http_port
        self.http_host = http_host
        self._proxy = None
        self._bg = None
        self._shutdown = False

    def _ensure_bg(self):
        if self._bg is None:
            loop = asyncio.get_event_loop()
            self._bg = loop.create_task(self._reconcile_loop())
            loop.create_task(self._recover_state_from_checkpoint())

    async def _graceful_stop_replica(self, h, timeout_s: float):
        """Drain in-flight requests before killing a removed replica
        (parity: reference graceful_shutdown_timeout_s). The replica is
        already out of the routing set, so its queue only shrinks."""
        deadline = asyncio.get_event_loop().time() + max(timeout_s, 0.0)
        while asyncio.get_event_loop().time() < deadline:
            try:
                if await _aw(h.num_ongoing_requests.remote()) == 0:
                    break
            except Exception:
                break
            await asyncio.sleep(0.2)
        try:
            ray.kill(h)
        except Exception:
            pass

    # ------------------------------------------------- crash recovery
    def _checkpoint_state(self):
        """Persist the app configs to the GCS KV (parity: the reference
        controller's KV checkpoint, controller.py:586) so a restarted
        controller (max_restarts=-1) redeploys everything."""
        try:
            snap = pickle.dumps({
                "apps": {
                    app: {
                        "route_prefix": self.route_prefixes.get(app),
                        "ingress": self.ingress.get(app),
                        "deployments": [dr.cfg for dr in deps.values()],
                    }
                    for app, deps in self.apps.items()
                },
                "http": (self.http_host, self.http_port),
            })
            from ant_ray_amd.experimental import internal_kv

            internal_kv._internal_kv_put(b"controller_ckpt", snap,
                                         namespace=b"serve")
        except Exception:
            logger.exception("serve controller checkpoint failed")

    async def _recover_state_from_checkpoint(self):
        if self.apps:
            return  # fresh deploys already arrived
        try:
            from ant_ray_amd.experimental import internal_kv

            raw = internal_kv._internal_kv_get(b"controller_ckpt",
                                               namespace=b"serve")
        except Exception:
            return
        if not raw:
            return
        try:
            snap = pickle.loads(raw)
        except Exception:
            logger.exception("bad serve controller checkpoint")
            return
        for app, info in snap.get("apps", {}).items():
            if app in self.apps:
                continue
            logger.warning("serve controller: recovering app %r from "
                           "checkpoint", app)
            try:
                await self.deploy_application(
                    app, info.get("route_prefix") or "/",
                    info.get("deployments") or [],
                    info.get("ingress"))
            except Exception:
                logger.exception("recovery of app %r failed", app)

    # ------------------------------------------------------------- deploy

    async def deploy_application(self, name: str, route_prefix: str,
                                 deployments: List[dict], ingress: str):
        """deployments: [{name, callable_bytes, init_args, init_kwargs,
        num_replicas, max_ongoing_requests, ray_actor_options,
        autoscaling_config, user_config}]"""
        self._ensure_bg()
        app = self.apps.setdefault(name, {})
        wanted = {d["name"] for d in deployments}
        for dep_name in list(app):
            if dep_name not in wanted:
                await self._scale_to(app[dep_name], 0, name, dep_name)
                del app[dep_name]
        for d in deployments:
            dr = app.setdefault(d["name"], DeploymentReplicas())
            dr.cfg = d
            auto = AutoscalingConfig.coerce(d.get("autoscaling_config"))
            dr.cfg["autoscaling_config"] = auto
            if auto is not None:
                target = max(auto.min_replicas, min(auto.max_replicas,
                                                    dr.target or auto.min_replicas))
            else:
                target = d.get("num_replicas", 1)
            await self._scale_to(dr, target, name, d["name"])
        self.ingress[name] = ingress
        self.route_prefixes[name] = route_prefix
        self._checkpoint_state()
        return True

    async def delete_application(self, name: str):
        app = self.apps.pop(name, {})
        for dep_name, dr in app.items():
            await self._scale_to(dr, 0, name, dep_name)
        self.ingress.pop(name, None)
        self.route_prefixes.pop(name, None)
        self._checkpoint_state()
        return True

    async def _scale_to(self, dr: DeploymentReplicas, target: int,
                        app: str, dep_name: str):
        dr.target = target
        d = dr.cfg
        while len(dr.replicas) < target:
            opts = dict(d.get("ray_actor_options") or {})
            opts.setdefault("num_cpus", 1)
            opts["max_concurrency"] = max(d.get("max_ongoing_requests", 100), 8)
            ReplicaCls = ray.remote(Replica)
            rank = len(dr.replicas)
            h = ReplicaCls.options(**opts).remote(
                d["callable_bytes"], d.get("init_args") or (),
                d.get("init_kwargs") or {}, d.get("user_config"),
                context={"app_name": app, "deployment": dep_name,
                         "replica_tag": f"{app}#{dep_name}#{rank}",
                         "rank": rank, "world_size": target},
            )
            dr.replicas.append(h)
            logger.info("started replica %d of %s/%s", len(dr.replicas), app,
                        dep_name)
        while len(dr.replicas) > target:
            h = dr.replicas.pop()
            asyncio.get_event_loop().create_task(
                self._graceful_stop_replica(
                    h, float(d.get("graceful_shutdown_timeout_s", 20.0)
                             if isinstance(d, dict) else 20.0)))
        # wait until new replicas construct (first health check) — bounded:
        # an unplaceable replica (cluster out of CPUs/GPUs) must surface as
        # a deploy error, not an indefinite hang (parity: the reference's
        # deploy timeout / "1 replica pending allocation" status)
        if dr.replicas:
            import os

            deploy_timeout = float(
                os.environ.get("ANTRAY_SERVE_DEPLOY_TIMEOUT_S", "60"))
            try:
                await asyncio.wait_for(
                    asyncio.gather(*[
                        _aw(h.check_health.remote()) for h in dr.replicas
                    ], return_exceptions=True),
                    timeout=deploy_timeout)
            except asyncio.TimeoutError:
                raise RuntimeError(
                    f"deployment {app}/{dep_name}: replicas not ready after "
                    f"{deploy_timeout:.0f}s — likely pending allocation "
                    "(insufficient cluster resources for "
                    f"{len(dr.replicas)} replicas)") from None

    async def getpid(self) -> int:
        import os

        self._ensure_bg()
        return os.getpid()

    # ------------------------------------------------------------ queries

    async def get_replicas(self, app: str, deployment: str):
        dr = self.apps.get(app, {}).get(deployment)
        if dr is None:
            return None
        return list(dr.replicas)

    async def get_deployment_info(self, app: str, deployment: str):
        dr = self.apps.get(app, {}).get(deployment)
        if dr is None:
            return None
        return {"target": dr.target, "num_replicas": len(dr.replicas),
                "max_ongoing_requests": dr.cfg.get("max_ongoing_requests")}

    async def get_app_config(self, app: str):
        if app not in self.apps:
            return None
        return {"ingress": self.ingress.get(app),
                "route_prefix": self.route_prefixes.get(app),
                "deployments": list(self.apps[app])}

    async def list_applications(self):
        self._ensure_bg()  # first call after a restart triggers recovery
        def _ing_streaming(app, deps):
            ing = self.ingress.get(app)
            dr = deps.get(ing) if ing else None
            return bool(dr and dr.cfg.get("is_streaming"))

        return {
            app: {
                "route_prefix": self.route_prefixes.get(app),
                "ingress": self.ingress.get(app),
                "ingress_streaming": _ing_streaming(app, deps),
                "deployments": {
                    name: {"replicas": len(dr.replicas), "target": dr.target}
                    for name, dr in deps.items()
                },
            }
            for app, deps in self.apps.items()
        }

    async def graceful_shutdown(self):
        self._shutdown = True
        for name in list(self.apps):
            await self.delete_application(name)
        return True

    # -------------------------------------------------- reconcile/autoscale

    async def _reconcile_loop(self):
        while not self._shutdown:
            try:
                await self._reconcile_once()
            except Exception:
                logger.exception("reconcile failed")
            await asyncio.sleep(1.0)

    async def _reconcile_once(self):
        for app, deps in self.apps.items():
            for dep_name, dr in deps.items():
                # liveness + periodic USER health check (reference
                # health_check_period_s: an unhealthy replica — raise or
                # falsy return from check_health — is replaced like a
                # dead one)
                period = float(dr.cfg.get("health_check_period_s", 10.0))
                due = (time.monotonic()
                       - getattr(dr, "last_health_check", 0.0)) >= period
                if due:
                    dr.last_health_check = time.monotonic()
                alive = []
                for h in dr.replicas:
                    try:
                        if due:
                            healthy = await _aw(h.check_health.remote())
                            if healthy is False:
                                raise RuntimeError("check_health falsy")
                        else:
                            await _aw(h.num_ongoing_requests.remote())
                        alive.append(h)
                    except Exception:
                        logger.warning(
                            "replica of %s/%s dead or unhealthy; replacing",
                            app, dep_name)
                        try:
                            ray.kill(h)
                        except Exception:
                            pass
                dr.replicas = alive
                if len(dr.replicas) < dr.target:
                    await self._scale_to(dr, dr.target, app, dep_name)
                auto: Optional[AutoscalingConfig] = dr.cfg.get("autoscaling_config")
                if auto is None or not dr.replicas:
                    continue
                counts = await asyncio.gather(*[
                    _aw(h.num_ongoing_requests.remote()) for h in dr.replicas
                ], return_exceptions=True)
                total = sum(c for c in counts if isinstance(c, int))
                desired = _calculate_desired_num_replicas(auto, total,
                                                          len(dr.replicas))
                now = time.monotonic()
                if desired > dr.target and now - dr.last_scale_up > auto.upscale_delay_s:
                    dr.last_scale_up = now
                    await self._scale_to(dr, desired, app, dep_name)
                elif (desired < dr.target
                      and now - dr.last_scale_down > auto.downscale_delay_s):
                    dr.last_scale_down = now
                    await self._scale_to(dr, desired, app, dep_name)


async def _aw(ref):
    """Await an ObjectRef inside the controller's event loop."""
    return await ref
