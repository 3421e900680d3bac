"""SIGUSR1 → all-thread stack dump (backs the `ray stack` CLI).

Role parity: reference `ray stack` uses py-spy to attach; py-spy is not
in this image, so every runtime process registers faulthandler on
SIGUSR1 at startup and `ray stack` signals them — the dumps land on each
process's stderr, which the session log files capture.
"""
import faulthandler
import signal


def install():
    try:
        faulthandler.register(signal.SIGUSR1, all_threads=True, chain=True)
    except (AttributeError, ValueError, OSError):
        pass  # non-main thread / unsupported platform
