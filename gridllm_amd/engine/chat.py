"""Chat templating (Llama-3 header style).

The reference's Ollama path flattens chat to "role: content" text
(reference: server/src/routes/ollama.ts:366-370) while its OpenAI path does
real chat (openai.ts:651-654). We template properly on BOTH paths
(SURVEY.md §7 hard-part 6 recommendation) while keeping wire shapes intact.
"""

from __future__ import annotations

from typing import Iterable


def render_chat(messages: Iterable[dict]) -> str:
    parts = ["<|begin_of_text|>"]
    for m in messages:
        role = m.get("role", "user")
        content = m.get("content", "")
        if isinstance(content, list):  # OpenAI multimodal split: text parts only
            content = "\n".join(p.get("text", "") for p in content
                                if isinstance(p, dict) and p.get("type") == "text")
        parts.append(f"<|start_header_id|>{role}<|end_header_id|>\n\n{content}<|eot_id|>")
    parts.append("<|start_header_id|>assistant<|end_header_id|>\n\n")
    return "".join(parts)


CHAT_STOPS = ["<|eot_id|>", "<|start_header_id|>"]
