"""Model architecture registry.

The reference has no model definitions at all (compute lived in Ollama);
these specs reproduce the Llama-family architectures its deployments serve
via Ollama model tags (BASELINE.json configs: llama3.2:1b, llama3.2:3b,
llama3:70b).
"""

from __future__ import annotations

from dataclasses import dataclass, replace


@dataclass(frozen=True)
class ModelSpec:
    name: str
    hidden: int
    layers: int
    heads: int
    kv_heads: int
    head_dim: int
    intermediate: int
    vocab: int
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    max_seq_len: int = 8192
    tie_embeddings: bool = True
    qkv_bias: bool = False         # Qwen2-family: bias on q/k/v projections

    @property
    def params(self) -> int:
        h, f, v = self.hidden, self.intermediate, self.vocab
        qkv = (self.heads + 2 * self.kv_heads) * self.head_dim * h
        o = self.heads * self.head_dim * h
        mlp = 3 * h * f
        if self.qkv_bias:
            qkv += (self.heads + 2 * self.kv_heads) * self.head_dim
        per_layer = qkv + o + mlp + 2 * h
        emb = v * h * (1 if self.tie_embeddings else 2)
        return self.layers * per_layer + emb + h


_SPECS: dict[str, ModelSpec] = {}


def _reg(spec: ModelSpec, *aliases: str) -> None:
    _SPECS[spec.name] = spec
    for a in aliases:
        _SPECS[a] = spec


_reg(ModelSpec("llama3.2:1b", hidden=2048, layers=16, heads=32, kv_heads=8,
               head_dim=64, intermediate=8192, vocab=128256),
     "llama3.2-1b")
_reg(ModelSpec("llama3.2:3b", hidden=3072, layers=28, heads=24, kv_heads=8,
               head_dim=128, intermediate=8192, vocab=128256),
     "llama3.2", "llama3.2:latest", "llama3.2-3b")
_reg(ModelSpec("llama3:8b", hidden=4096, layers=32, heads=32, kv_heads=8,
               head_dim=128, intermediate=14336, vocab=128256,
               tie_embeddings=False),
     "llama3.1:8b", "llama3-8b", "llama3", "llama3:latest")
_reg(ModelSpec("llama3:70b", hidden=8192, layers=80, heads=64, kv_heads=8,
               head_dim=128, intermediate=28672, vocab=128256,
               tie_embeddings=False),
     "llama3.1:70b", "llama3-70b")
# Qwen2.5 family (the reference's own CI model is qwen2.5:0.5b,
# .github/workflows/integration-tests.yml:28): llama-style GQA +
# SwiGLU with q/k/v projection BIAS and rope theta 1e6
_reg(ModelSpec("qwen2.5:0.5b", hidden=896, layers=24, heads=14, kv_heads=2,
               head_dim=64, intermediate=4864, vocab=151936,
               rope_theta=1e6, norm_eps=1e-6, max_seq_len=32768,
               qkv_bias=True),
     "qwen2.5-0.5b", "qwen2.5:0.5b-instruct")
_reg(ModelSpec("qwen2.5:1.5b", hidden=1536, layers=28, heads=12, kv_heads=2,
               head_dim=128, intermediate=8960, vocab=151936,
               rope_theta=1e6, norm_eps=1e-6, max_seq_len=32768,
               qkv_bias=True),
     "qwen2.5-1.5b")
_reg(ModelSpec("qwen2.5:7b", hidden=3584, layers=28, heads=28, kv_heads=4,
               head_dim=128, intermediate=18944, vocab=152064,
               rope_theta=1e6, norm_eps=1e-6, max_seq_len=32768,
               qkv_bias=True, tie_embeddings=False),
     "qwen2.5-7b", "qwen2.5", "qwen2.5:latest")
# tiny configs for CPU tests and CI (tiny2: a SECOND architecture so
# mixed-fleet routing tests have two distinct models on CPU)
_reg(ModelSpec("tiny", hidden=256, layers=2, heads=4, kv_heads=2, head_dim=64,
               intermediate=512, vocab=512, max_seq_len=256),
     "tiny:latest")
_reg(ModelSpec("tiny2", hidden=128, layers=2, heads=2, kv_heads=2,
               head_dim=64, intermediate=256, vocab=512, max_seq_len=256),
     "tiny2:latest")
_reg(ModelSpec("tiny-qwen", hidden=128, layers=2, heads=4, kv_heads=2,
               head_dim=32, intermediate=256, vocab=512, max_seq_len=256,
               rope_theta=1e6, norm_eps=1e-6, qkv_bias=True),
     "tiny-qwen:latest")


def get_spec(name: str, max_seq_len: int = 0) -> ModelSpec:
    key = name.strip()
    if key not in _SPECS:
        raise KeyError(f"unknown model {name!r}; known: {sorted(set(_SPECS))}")
    spec = _SPECS[key]
    if max_seq_len and max_seq_len != spec.max_seq_len:
        spec = replace(spec, max_seq_len=max_seq_len)
    return spec


def list_specs() -> list[str]:
    return sorted({s.name for s in _SPECS.values()})


def register_spec(spec: ModelSpec, *aliases: str) -> None:
    """Register a custom architecture at runtime (tests, custom configs)."""
    _reg(spec, *aliases)


def names_for(name: str) -> list[str]:
    """Canonical name + every alias of a model — what a worker should
    advertise so clients can use any Ollama-style spelling (the reference
    advertised Ollama's full tag list, WorkerClientService.ts:129-154)."""
    spec = get_spec(name)
    return sorted({k for k, v in _SPECS.items() if v.name == spec.name})
