"""ray.cross_language (parity: reference cross_language.py). Java/C++
workers are not part of this MI355X build; see PARITY.md §2.2."""
from ant_ray_amd import (  # noqa: F401
    cpp_function,
    java_actor_class,
    java_function,
)
