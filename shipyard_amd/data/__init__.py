"""Data plane: object store, SYSHARD shard format, integrity manifests,
parallel mover, NVMe->HBM stager, storage clusters, and the pure-python
LZ4 reference codec (see SURVEY.md §2.5 hot paths)."""

from .storage import ObjectStore  # noqa: F401
