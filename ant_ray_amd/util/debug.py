"""log_once / periodic logging helpers.

Role parity: reference python/ray/util/debug.py (log_once:18,
disable_log_once_globally, enable_periodic_logging).
"""
import time

_logged = set()
_disabled = False
_periodic_log = False
_last_logged = 0.0


def log_once(key) -> bool:
    """True only the first time this key is seen in this process (or once
    per 60 s after enable_periodic_logging())."""
    global _last_logged
    if _disabled:
        return False
    if key not in _logged:
        _logged.add(key)
        _last_logged = time.time()
        return True
    if _periodic_log and time.time() - _last_logged > 60.0:
        _logged.clear()
        _last_logged = time.time()
        return False
    return False


def disable_log_once_globally():
    """Make log_once() return False in this process."""
    global _disabled
    _disabled = True


def enable_periodic_logging():
    """Make log_once() periodically reset its seen-key set."""
    global _periodic_log
    _periodic_log = True


def reset_log_once(key):
    _logged.discard(key)
