"""GridLLM-AMD: an MI355X-native distributed LLM serving framework.

A from-scratch rebuild of the capabilities of GridLLM (reference:
server/src + client/src TypeScript tree): a job scheduler + worker
registry + Ollama/OpenAI-compatible gateway, with the inference engine
implemented in-process on PyTorch-ROCm + hand-written CDNA4 (gfx950)
HIP kernels instead of delegating to an external Ollama daemon.

Layers:
  L5 gateway     gridllm_amd.server.routes   (Ollama + OpenAI + native API)
  L4 orchestration gridllm_amd.server        (JobScheduler, WorkerRegistry)
  L3 message bus gridllm_amd.bus             (asyncio TCP pub/sub + KV; the
                                              reference used Redis — this image
                                              has none, so the bus is built in)
  L2 worker      gridllm_amd.worker          (WorkerAgent)
  L1 engine      gridllm_amd.engine          (the Ollama replacement:
                                              HIP/CDNA4 kernels, hipGraph decode)
"""

__version__ = "0.1.0"
