from gridllm_amd.parallel.tp import TPContext  # noqa: F401
