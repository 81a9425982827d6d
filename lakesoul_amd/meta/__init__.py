from .entities import (  # noqa: F401
    CommitOp,
    DataCommitInfo,
    DataFileOp,
    FileOp,
    Namespace,
    PartitionInfo,
    TableInfo,
)
from .client import MetaClient  # noqa: F401
