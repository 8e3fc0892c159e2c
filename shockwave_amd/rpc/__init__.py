from .services import (
    IteratorRpcClient,
    SchedulerRpcClient,
    WorkerRpcClient,
    serve_scheduler,
    serve_worker,
)
from .transport import RpcClient, make_server

__all__ = [
    "IteratorRpcClient",
    "SchedulerRpcClient",
    "WorkerRpcClient",
    "serve_scheduler",
    "serve_worker",
    "RpcClient",
    "make_server",
]
