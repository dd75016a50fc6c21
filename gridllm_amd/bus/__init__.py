from gridllm_amd.bus.base import Bus  # noqa: F401
from gridllm_amd.bus.memory import MemoryBusCore, MemoryBus  # noqa: F401
from gridllm_amd.bus.tcp import BusServer, TcpBus  # noqa: F401
