from gridllm_amd.worker.agent import WorkerAgent  # noqa: F401
