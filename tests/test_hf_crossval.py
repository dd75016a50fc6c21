"""Cross-validation against the canonical HuggingFace implementations
(transformers is installed from the offline wheelhouse): our model math
— fused QKV/gate-up layout, RMSNorm, rotate-half RoPE, GQA attention,
SwiGLU, tied lm_head, Qwen2's qkv bias — must reproduce HF Llama/Qwen2
logits on the same random weights. This validates the ENGINE's
architecture implementation independently of our own CPU reference ops.
(Reference parity context: the reference's CI model is qwen2.5:0.5b,
.github/workflows/integration-tests.yml:28 — compute lived in Ollama.)"""

import pytest
import torch

pytest.importorskip("transformers")


def _build_ours(name: str):
    from gridllm_amd.models.llama import LlamaModel
    from gridllm_amd.models.registry import get_spec
    spec = get_spec(name, max_seq_len=128)
    return LlamaModel(spec, device="cpu", dtype="fp32", kv_slots=2, seed=7)


def _to_hf(ours, qkv_bias: bool):
    """Copy our fused weights into a matching HF model."""
    import transformers
    s = ours.spec
    d = s.head_dim
    hq, hkv, f = s.heads, s.kv_heads, s.intermediate
    common = dict(
        vocab_size=s.vocab, hidden_size=s.hidden,
        intermediate_size=f, num_hidden_layers=s.layers,
        num_attention_heads=hq, num_key_value_heads=hkv,
        rms_norm_eps=s.norm_eps, rope_theta=s.rope_theta,
        max_position_embeddings=128, tie_word_embeddings=True,
        attention_dropout=0.0)
    if qkv_bias:
        cfg = transformers.Qwen2Config(**common)
        hf = transformers.Qwen2ForCausalLM(cfg)
    else:
        cfg = transformers.LlamaConfig(
            **common, attention_bias=False, mlp_bias=False)
        hf = transformers.LlamaForCausalLM(cfg)
    hf.eval()
    sd = hf.state_dict()
    new = {}
    new["model.embed_tokens.weight"] = ours.embed
    new["model.norm.weight"] = ours.final_norm
    if "lm_head.weight" in sd:
        new["lm_head.weight"] = ours.lm_head
    for i, lw in enumerate(ours.layers):
        p = f"model.layers.{i}."
        new[p + "self_attn.q_proj.weight"] = lw.wqkv[:hq * d]
        new[p + "self_attn.k_proj.weight"] = lw.wqkv[hq * d:(hq + hkv) * d]
        new[p + "self_attn.v_proj.weight"] = lw.wqkv[(hq + hkv) * d:]
        new[p + "self_attn.o_proj.weight"] = lw.wo
        new[p + "mlp.gate_proj.weight"] = lw.wgateup[:f]
        new[p + "mlp.up_proj.weight"] = lw.wgateup[f:]
        new[p + "mlp.down_proj.weight"] = lw.wdown
        new[p + "input_layernorm.weight"] = lw.attn_norm
        new[p + "post_attention_layernorm.weight"] = lw.mlp_norm
        if qkv_bias:
            new[p + "self_attn.q_proj.bias"] = lw.bqkv[:hq * d]
            new[p + "self_attn.k_proj.bias"] = lw.bqkv[hq * d:(hq + hkv) * d]
            new[p + "self_attn.v_proj.bias"] = lw.bqkv[(hq + hkv) * d:]
    missing = set(sd) - set(new)
    assert not missing, f"unmapped HF params: {sorted(missing)[:6]}"
    hf.load_state_dict({k: v.clone().float() for k, v in new.items()},
                       strict=True)
    return hf


@torch.no_grad()
def _compare(name: str, qkv_bias: bool):
    ours = _build_ours(name)
    hf = _to_hf(ours, qkv_bias)
    v = ours.spec.vocab
    ids = [(37 * (i + 2)) % (v - 4) for i in range(9)]
    # prefill: our last-position logits vs HF
    slot = ours.cache.alloc()
    logits = ours.forward_prefill(torch.tensor(ids, dtype=torch.int64), slot)
    ref = hf(input_ids=torch.tensor([ids])).logits[0]
    torch.testing.assert_close(logits.float(), ref[-1].float(),
                               atol=2e-3, rtol=2e-3)
    # forced decode steps vs HF full-sequence recompute
    seq = list(ids)
    for step in range(4):
        t = (911 * (step + 1)) % (v - 4)
        dl = ours.forward_decode(
            torch.tensor([t], dtype=torch.int64),
            torch.tensor([slot], dtype=torch.int32))[0]
        seq.append(t)
        ref = hf(input_ids=torch.tensor([seq])).logits[0, -1]
        torch.testing.assert_close(dl.float(), ref.float(),
                                   atol=2e-3, rtol=2e-3)


def test_llama_family_matches_transformers():
    _compare("tiny", qkv_bias=False)


def test_qwen2_family_matches_transformers():
    _compare("tiny-qwen", qkv_bias=True)
