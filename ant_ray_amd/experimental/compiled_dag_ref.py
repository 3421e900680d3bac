"""ray.experimental.compiled_dag_ref (parity: reference module of the
same name — CompiledDAGRef lives with the DAG implementation here)."""
from ant_ray_amd.dag.node import CompiledDAGRef  # noqa: F401
