from gridllm_amd.server.registry import WorkerRegistry  # noqa: F401
from gridllm_amd.server.scheduler import JobScheduler  # noqa: F401
