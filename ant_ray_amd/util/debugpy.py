"""ray.util.ray_debugpy — debugpy attach points.

Role parity: reference python/ray/util/debugpy.py. The `debugpy` package
is not in this air-gapped image, so set_trace() logs and continues
(matching the reference's behavior when debugpy is unimportable).
"""
import logging

logger = logging.getLogger(__name__)


def set_trace(breakpoint_uuid=None):
    try:
        import debugpy  # noqa: F401
    except ImportError:
        logger.warning(
            "ray.util.ray_debugpy.set_trace(): the `debugpy` package is not "
            "installed in this image; breakpoint skipped.")
        return
    debugpy.listen(("127.0.0.1", 0))
    debugpy.wait_for_client()
    debugpy.breakpoint()
