from gridllm_amd.engine.engine import InferenceEngine, GenerationResult  # noqa: F401
