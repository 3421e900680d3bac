"""ray.util.inspect_serializability parity: explain WHY an object fails
to serialize (walks closures/attributes reporting offending members)."""
from __future__ import annotations

from typing import Any, Set, Tuple


def inspect_serializability(obj: Any, name: str = None,
                            depth: int = 3) -> Tuple[bool, Set[str]]:
    """Returns (serializable, failure_set). Prints a short trace of the
    first offending members, like the reference helper."""
    from ant_ray_amd._private import serialization

    name = name or getattr(obj, "__name__", type(obj).__name__)
    failures: Set[str] = set()
    ok = _check(obj, name, depth, failures)
    if ok:
        print(f"{name!r} is serializable")
    else:
        print(f"{name!r} is NOT serializable; offending members:")
        for f in sorted(failures):
            print(f"  - {f}")
    return ok, failures


def _try(obj) -> bool:
    from ant_ray_amd._private import serialization

    try:
        serialization.dumps_by_value(obj)
        return True
    except Exception:
        return False


def _check(obj, path: str, depth: int, failures) -> bool:
    if _try(obj):
        return True
    if depth <= 0:
        failures.add(path)
        return False
    found_cause = False
    closure = getattr(obj, "__closure__", None)
    if closure:
        names = obj.__code__.co_freevars
        for nm, cell in zip(names, closure):
            try:
                inner = cell.cell_contents
            except ValueError:
                continue
            if not _try(inner):
                found_cause = True
                _check(inner, f"{path}.<closure>.{nm}", depth - 1, failures)
    code = getattr(obj, "__code__", None)
    gl = getattr(obj, "__globals__", None)
    if code is not None and isinstance(gl, dict):
        for nm in code.co_names:
            if nm in gl and not _try(gl[nm]):
                found_cause = True
                _check(gl[nm], f"{path}.<global>.{nm}", depth - 1, failures)
    d = getattr(obj, "__dict__", None)
    if isinstance(d, dict):
        for k, v in list(d.items())[:64]:
            if not _try(v):
                found_cause = True
                _check(v, f"{path}.{k}", depth - 1, failures)
    if not found_cause:
        failures.add(path)
    return False
