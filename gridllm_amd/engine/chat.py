"""Chat templating, per model family.

The reference's Ollama path flattens chat to "role: content" text
(reference: server/src/routes/ollama.ts:366-370) while its OpenAI path does
real chat (openai.ts:651-654). We template properly on BOTH paths
(SURVEY.md §7 hard-part 6 recommendation) while keeping wire shapes intact.
Families: Llama-3 header style (default) and ChatML (Qwen2.5 — the
reference's CI model family)."""

from __future__ import annotations

from typing import Iterable


def _content_text(m: dict) -> str:
    content = m.get("content", "")
    if isinstance(content, list):  # OpenAI multimodal split: text parts only
        content = "\n".join(p.get("text", "") for p in content
                            if isinstance(p, dict) and p.get("type") == "text")
    return content


def family_of(model_name: str) -> str:
    n = (model_name or "").lower()
    return "qwen" if "qwen" in n else "llama"


def render_chat(messages: Iterable[dict], think: bool = False,
                family: str = "llama") -> str:
    if family == "qwen":
        # ChatML (Qwen2/2.5): <|im_start|>role\ncontent<|im_end|>
        parts = []
        for m in messages:
            parts.append(f"<|im_start|>{m.get('role', 'user')}\n"
                         f"{_content_text(m)}<|im_end|>\n")
        parts.append("<|im_start|>assistant\n")
        if think:
            parts.append(THINK_OPEN + "\n")
        return "".join(parts)
    parts = ["<|begin_of_text|>"]
    for m in messages:
        role = m.get("role", "user")
        parts.append(f"<|start_header_id|>{role}<|end_header_id|>"
                     f"\n\n{_content_text(m)}<|eot_id|>")
    parts.append("<|start_header_id|>assistant<|end_header_id|>\n\n")
    if think:
        # thinking mode (Ollama `think: true`, reference forwards it at
        # server/src/routes/ollama.ts:216-239): generation starts inside a
        # think block; output before THINK_CLOSE is the `thinking` field
        parts.append(THINK_OPEN + "\n")
    return "".join(parts)


CHAT_STOPS = ["<|eot_id|>", "<|start_header_id|>"]
_CHAT_STOPS_BY_FAMILY = {"llama": CHAT_STOPS,
                         "qwen": ["<|im_end|>", "<|im_start|>"]}


def chat_stops(family: str = "llama") -> list:
    return list(_CHAT_STOPS_BY_FAMILY.get(family, CHAT_STOPS))
THINK_OPEN = "<think>"
THINK_CLOSE = "</think>"


class ThinkSplitter:
    """Incremental splitter for thinking-mode output: text before the
    first THINK_CLOSE is `thinking`, the rest is `content`. Streaming-safe
    (holds back a partial close tag across piece boundaries)."""

    def __init__(self) -> None:
        self.in_think = True
        self._tail = ""

    def push(self, piece: str) -> tuple[str, str]:
        """Returns (thinking_delta, content_delta) ready to emit."""
        if not self.in_think:
            return "", piece
        buf = self._tail + piece
        idx = buf.find(THINK_CLOSE)
        if idx >= 0:
            self.in_think = False
            self._tail = ""
            content = buf[idx + len(THINK_CLOSE):].lstrip("\n")
            return buf[:idx], content
        # hold back the longest buf suffix that could begin THINK_CLOSE
        keep = 0
        for k in range(min(len(buf), len(THINK_CLOSE) - 1), 0, -1):
            if THINK_CLOSE.startswith(buf[-k:]):
                keep = k
                break
        self._tail = buf[len(buf) - keep:] if keep else ""
        return (buf[:len(buf) - keep] if keep else buf), ""

    def flush(self) -> tuple[str, str]:
        t, self._tail = self._tail, ""
        if self.in_think:
            return t, ""
        return "", t


def split_think(text: str) -> tuple[str, str]:
    """One-shot split of a completed thinking-mode generation."""
    idx = text.find(THINK_CLOSE)
    if idx < 0:
        return text, ""
    return text[:idx], text[idx + len(THINK_CLOSE):].lstrip("\n")
