from .comms import (
    Comms,
    TorchDistComms,
    LoopbackComms,
    ReduceOp,
    init as init_comms,
    inject_comms,
)

__all__ = ["Comms", "TorchDistComms", "LoopbackComms", "ReduceOp", "init_comms", "inject_comms"]
