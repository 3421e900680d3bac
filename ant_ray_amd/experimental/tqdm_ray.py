"""Distributed-safe tqdm.

Role parity: reference python/ray/experimental/tqdm_ray.py (tqdm:55,
safe_print:37, instance:384). The reference multiplexes worker progress
bars onto the driver terminal via magic-token log lines; in this build
each process writes its own bar straight to stderr (worker stderr lands
in the session logs, the driver renders normally), which keeps the API —
tqdm/update/set_description/close, safe_print — without the log-routing
daemon.
"""
import sys
import threading
from typing import Any, Iterable, Optional

_print_lock = threading.RLock()


def safe_print(*args, **kwargs):
    """Print without corrupting an active progress bar line."""
    with _print_lock:
        sys.stderr.write("\r\033[K")
        print(*args, **kwargs)


class tqdm:
    """API-compatible progress bar; wraps the real tqdm when available."""

    def __init__(self, iterable: Optional[Iterable] = None,
                 desc: Optional[str] = None, total: Optional[int] = None,
                 position: Optional[int] = None, flush_interval_s: float = 0.1,
                 **kwargs):
        self._iterable = iterable
        try:
            import tqdm.auto as _real

            self._bar = _real.tqdm(iterable=iterable, desc=desc, total=total,
                                   position=position, file=sys.stderr,
                                   **kwargs)
        except Exception:
            self._bar = None
            self._n = 0
            self._desc = desc or ""
            self._total = total

    def __iter__(self):
        if self._bar is not None:
            return iter(self._bar)

        def gen():
            for x in self._iterable:
                self.update(1)
                yield x

        return gen()

    def update(self, n: int = 1):
        if self._bar is not None:
            self._bar.update(n)
        else:
            self._n += n
            sys.stderr.write(f"\r{self._desc}: {self._n}"
                             + (f"/{self._total}" if self._total else ""))

    def set_description(self, desc: str):
        if self._bar is not None:
            self._bar.set_description(desc)
        else:
            self._desc = desc

    def refresh(self):
        if self._bar is not None:
            self._bar.refresh()

    def close(self):
        if self._bar is not None:
            self._bar.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class _BarManager:
    def unhide_bars(self):
        pass


_instance = _BarManager()


def instance() -> _BarManager:
    return _instance
