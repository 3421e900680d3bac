"""Accelerator managers — AMD-first.

Role parity: reference python/ray/_private/accelerators/ (per-vendor
managers; amd_gpu.py:33 AMDGPUAcceleratorManager with HIP_VISIBLE_DEVICES
at :10). This build targets MI355X: the AMD manager is the real one (its
detection feeds the raylet's GPU pool); the NVIDIA manager exists only to
report zero devices on this platform.
"""
from __future__ import annotations

import os
from typing import List, Optional


class AcceleratorManager:
    @staticmethod
    def get_resource_name() -> str:
        return "GPU"

    @staticmethod
    def get_current_node_num_accelerators() -> int:
        raise NotImplementedError

    @staticmethod
    def get_visible_accelerator_ids_env_var() -> str:
        raise NotImplementedError

    @staticmethod
    def set_visible_accelerator_ids(ids: List[int]) -> None:
        raise NotImplementedError


class AMDGPUAcceleratorManager(AcceleratorManager):
    """Parity: accelerators/amd_gpu.py:33."""

    VISIBLE_ENV = "HIP_VISIBLE_DEVICES"

    @staticmethod
    def get_visible_accelerator_ids_env_var() -> str:
        return AMDGPUAcceleratorManager.VISIBLE_ENV

    @staticmethod
    def get_current_node_num_accelerators() -> int:
        from ant_ray_amd._private.raylet import detect_num_gpus

        return detect_num_gpus()

    @staticmethod
    def get_current_process_visible_accelerator_ids() -> Optional[List[str]]:
        v = os.environ.get(AMDGPUAcceleratorManager.VISIBLE_ENV)
        if v is None:
            return None
        return [] if v == "" else v.split(",")

    @staticmethod
    def set_visible_accelerator_ids(ids: List[int]) -> None:
        val = ",".join(str(i) for i in ids)
        os.environ[AMDGPUAcceleratorManager.VISIBLE_ENV] = val
        # torch-ROCm also honors CUDA_VISIBLE_DEVICES; keep them in sync
        os.environ["CUDA_VISIBLE_DEVICES"] = val

    @staticmethod
    def get_current_node_accelerator_type() -> Optional[str]:
        base = "/sys/class/kfd/kfd/topology/nodes"
        try:
            for d in sorted(os.listdir(base)):
                p = os.path.join(base, d, "properties")
                with open(p) as f:
                    props = dict(
                        line.split() for line in f if len(line.split()) == 2)
                if int(props.get("gfx_target_version", 0)) >= 90500:
                    return "MI355X"
        except Exception:
            pass
        return None


class NvidiaGPUAcceleratorManager(AcceleratorManager):
    """Present for API parity; this platform has no NVIDIA devices."""

    @staticmethod
    def get_visible_accelerator_ids_env_var() -> str:
        return "CUDA_VISIBLE_DEVICES"

    @staticmethod
    def get_current_node_num_accelerators() -> int:
        return 0


def get_accelerator_manager_for_resource(resource: str = "GPU"):
    return AMDGPUAcceleratorManager
