from .services import IteratorRpcClient

__all__ = ["IteratorRpcClient"]
