"""Function/tool-calling plumbing for the gateway.

The reference forwarded `tools` to Ollama's chat template and converted the
returned `tool_calls` into OpenAI shape (server/src/routes/openai.ts:246-304;
its integration test exercises tool definitions, tests/integration/
integration.ts:83-173). Our engine has no native tool grammar, so the
gateway renders tool definitions into a system instruction (JSON-call
convention), optionally constrains the output with the byte-level JSON
automaton (`format: "json"`) when a call is forced, and strictly parses a
whole-message JSON reply back into tool calls."""

from __future__ import annotations

import json
import uuid
from typing import Optional


def extract_functions(tools) -> list[dict]:
    """Normalize an OpenAI/Ollama `tools` array to bare function specs."""
    if not isinstance(tools, list):
        return []
    out = []
    for t in tools:
        if not isinstance(t, dict):
            continue
        if t.get("type", "function") != "function":
            continue
        fn = t.get("function", t)
        if isinstance(fn, dict) and fn.get("name"):
            out.append(fn)
    return out


def forced_call(tool_choice) -> bool:
    """True when the request REQUIRES a tool call (tool_choice 'required'
    or a specific function) — the gateway then sets format=json so the
    automaton guarantees a parseable reply."""
    if isinstance(tool_choice, dict):
        return bool((tool_choice.get("function") or {}).get("name"))
    return tool_choice == "required"


def system_message(fns: list[dict], tool_choice=None) -> dict:
    """System instruction carrying the tool specs + call convention."""
    forced = ""
    if isinstance(tool_choice, dict):
        name = (tool_choice.get("function") or {}).get("name")
        if name:
            forced = f" You MUST call the tool named {json.dumps(name)}."
    elif tool_choice == "required":
        forced = " You MUST call one of the tools."
    return {"role": "system", "content":
            "You can call tools. To call one, reply with ONLY a JSON "
            'object {"name": <tool name>, "arguments": <arguments '
            "object>} and nothing else." + forced +
            " Available tools: " + json.dumps(fns)}


def parse_calls(text: str) -> Optional[list[dict]]:
    """Strict whole-message parse: a JSON object {"name":…, "arguments":…}
    or a list of them. Returns [{"name": str, "arguments": dict}, …] or
    None when the text is a normal answer."""
    try:
        val = json.loads(text.strip())
    except Exception:  # noqa: BLE001
        return None
    calls = val if isinstance(val, list) else [val]
    out = []
    for c in calls:
        if not isinstance(c, dict) or not isinstance(c.get("name"), str):
            return None
        args = c.get("arguments", c.get("parameters", {}))
        if not isinstance(args, dict):
            return None
        out.append({"name": c["name"], "arguments": args})
    return out or None


def to_openai(calls: list[dict]) -> list[dict]:
    """OpenAI wire shape: arguments as a JSON STRING + call ids."""
    return [{"id": f"call_{uuid.uuid4().hex[:24]}", "type": "function",
             "function": {"name": c["name"],
                          "arguments": json.dumps(c["arguments"])}}
            for c in calls]


def to_ollama(calls: list[dict]) -> list[dict]:
    """Ollama wire shape: arguments stays an OBJECT."""
    return [{"function": {"name": c["name"], "arguments": c["arguments"]}}
            for c in calls]


def call_schema(fns: list[dict], tool_choice=None) -> dict:
    """JSON schema a FORCED tool call must satisfy: anyOf over the
    allowed functions, each branch pinning `name` to that function and
    validating `arguments` against ITS `parameters` — per-name dispatch,
    enforced by the schema automaton's parallel branches
    (engine/schema_constraints.py::_Union)."""
    names = [f["name"] for f in fns]
    if isinstance(tool_choice, dict):
        forced = (tool_choice.get("function") or {}).get("name")
        if forced in names:
            names = [forced]
    branches = []
    for f in fns:
        if f["name"] not in names:
            continue
        args = f.get("parameters")
        if not isinstance(args, dict) or not args:
            args = {"type": "object"}
        branches.append({"type": "object",
                         "properties": {"name": {"const": f["name"]},
                                        "arguments": args},
                         "required": ["name", "arguments"],
                         "additionalProperties": False})
    if len(branches) == 1:
        return branches[0]
    return {"anyOf": branches}
