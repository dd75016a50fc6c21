from gridllm_amd.models.registry import ModelSpec, get_spec, list_specs  # noqa: F401
