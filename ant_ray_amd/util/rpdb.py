"""ray.util.pdb — breakpoint helper inside tasks/actors.

Role parity: reference python/ray/util/rpdb.py (remote pdb over a
socket). Single-node-class build: workers share the head's terminal is
NOT guaranteed, so set_trace() degrades gracefully — it opens the
standard pdb when stdin is a TTY (local_mode / driver code) and logs a
breakpoint-skipped message otherwise instead of wedging a worker.
"""
import logging
import pdb as _pdb
import sys

logger = logging.getLogger(__name__)


def set_trace(breakpoint_uuid=None):
    if sys.stdin is not None and sys.stdin.isatty():
        _pdb.Pdb().set_trace(sys._getframe().f_back)
    else:
        logger.warning(
            "ray.util.pdb.set_trace(): stdin is not a TTY in this worker; "
            "breakpoint skipped (attach with ray.util.pdb inside "
            "local_mode, or use post-mortem on the raised error).")


def post_mortem():
    _pdb.post_mortem()
