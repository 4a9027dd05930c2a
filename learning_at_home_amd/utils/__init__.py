"""L0 utilities: serialization, framing, futures, nested structures."""

from .serializer import PytorchSerializer, MSGPackSerializer
from .nested import nested_flatten, nested_pack, nested_map, nested_compare
from .mpfuture import MPFuture
from .timed_storage import TimedStorage, DHTExpiration, get_dht_time
from .connection import send_message, recv_message, aread_message, awrite_message, Endpoint
from .logging import get_logger
from .tensor_descr import TensorDescriptor, BatchTensorDescriptor

__all__ = [
    "PytorchSerializer",
    "MSGPackSerializer",
    "nested_flatten",
    "nested_pack",
    "nested_map",
    "nested_compare",
    "MPFuture",
    "TimedStorage",
    "DHTExpiration",
    "get_dht_time",
    "send_message",
    "recv_message",
    "aread_message",
    "awrite_message",
    "Endpoint",
    "get_logger",
    "TensorDescriptor",
    "BatchTensorDescriptor",
]
