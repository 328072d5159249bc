"""Core runtime: resources/handle, torch-backed device arrays, serialization,
bitset, interruptible cancellation, logging.

Reference parity: raft/core/* (resources.hpp:39, device_resources.hpp:53,
handle.hpp:23, mdarray.hpp, numpy_serializer.hpp, bitset.hpp, interruptible.hpp).
"""
from .resources import (
    Resources,
    DeviceResources,
    Handle,
    DeviceResourcesSNMG,
    DeviceResourcesManager,
    get_resources,
)
from .device_ndarray import device_ndarray
from .serialize import serialize_mdspan, deserialize_mdspan, save_npy, load_npy
from .bitset import Bitset
from .interruptible import Interruptible, synchronize as interruptible_synchronize
from .logger import get_logger, set_level
from .trace import annotate, annotated
from .memory import MemoryStats, TrackingScope, ResourceMonitor, TemporaryDeviceBuffer
from .mr import (
    DeviceMemoryResource,
    TorchMemoryResource,
    PoolMemoryResource,
    LimitingAdaptor,
    TrackingAdaptor,
    MemoryLimitExceeded,
    WorkspaceBuffer,
    AllocationStats,
)
from .mdbuffer import MDBuffer, MemoryType, memory_type_dispatcher, copy_mdspan
from .error import RaftError, LogicError, HipError, expects, fail
from .kvp import KeyValuePair
from .span import host_span, device_span, subspan
from . import operators
from . import math

__all__ = [
    "RaftError", "LogicError", "HipError", "expects", "fail",
    "KeyValuePair", "operators", "math", "TemporaryDeviceBuffer",
    "Resources", "DeviceResources", "Handle", "DeviceResourcesSNMG",
    "DeviceResourcesManager", "get_resources", "device_ndarray",
    "serialize_mdspan", "deserialize_mdspan", "save_npy", "load_npy",
    "Bitset", "Interruptible", "interruptible_synchronize",
    "get_logger", "set_level", "annotate", "annotated",
    "MemoryStats", "TrackingScope", "ResourceMonitor",
    "DeviceMemoryResource", "TorchMemoryResource", "PoolMemoryResource",
    "LimitingAdaptor", "TrackingAdaptor", "MemoryLimitExceeded",
    "WorkspaceBuffer", "AllocationStats",
    "MDBuffer", "MemoryType", "memory_type_dispatcher", "copy_mdspan",
    "host_span", "device_span", "subspan",
]
