"""Local MI355X executor: the Azure Batch service seam, re-implemented
as an on-node scheduler over GPU slots."""

from .service import ExecutorError, LocalExecutor  # noqa: F401
from .store import Store  # noqa: F401
