"""Output post-processing parsers: tool calls + reasoning content.

Reference parity: the reference's postprocessor extracts tool calls and
reasoning segments from generated text
(ai-dynamo/dynamo lib/llm/src/preprocessor.rs:3953 tool-call parsing,
:4577 reasoning parsing). Two widely-used wire formats are handled:

  tool calls: `<tool_call>{"name": ..., "arguments": {...}}</tool_call>`
              (hermes style) or a bare top-level JSON object with
              "name" + "arguments" keys;
  reasoning:  `<think> ... </think>` segments split into
              `reasoning_content` (deepseek-r1 style).
"""
from __future__ import annotations

import json
import re
import uuid
from typing import List, Optional, Tuple

_TOOL_TAG = re.compile(r"<tool_call>\s*(\{.*?\})\s*</tool_call>", re.S)
_THINK_TAG = re.compile(r"<think>(.*?)</think>", re.S)


def parse_tool_calls(text: str) -> Tuple[str, List[dict]]:
    """Extract tool calls; returns (remaining_text, openai_tool_calls)."""
    calls: List[dict] = []

    def to_call(obj: dict) -> Optional[dict]:
        if not isinstance(obj, dict) or "name" not in obj:
            return None
        args = obj.get("arguments", obj.get("parameters", {}))
        if not isinstance(args, str):
            args = json.dumps(args)
        return {"id": f"call_{uuid.uuid4().hex[:24]}",
                "type": "function",
                "function": {"name": str(obj["name"]), "arguments": args}}

    def tag_sub(m):
        try:
            c = to_call(json.loads(m.group(1)))
        except json.JSONDecodeError:
            return m.group(0)
        if c is None:
            return m.group(0)
        calls.append(c)
        return ""

    rest = _TOOL_TAG.sub(tag_sub, text)

    if not calls:
        stripped = rest.strip()
        if stripped.startswith("{") and stripped.endswith("}"):
            try:
                c = to_call(json.loads(stripped))
                if c is not None:
                    calls.append(c)
                    rest = ""
            except json.JSONDecodeError:
                pass
    return rest, calls


def parse_reasoning(text: str) -> Tuple[str, Optional[str]]:
    """Split <think> segments; returns (content, reasoning_content)."""
    segs = _THINK_TAG.findall(text)
    if not segs:
        # unterminated opening tag: everything after it is reasoning
        if "<think>" in text:
            head, _, tail = text.partition("<think>")
            return head, tail.strip() or None
        return text, None
    rest = _THINK_TAG.sub("", text)
    return rest, "\n".join(s.strip() for s in segs if s.strip()) or None


def postprocess_chat(text: str):
    """Full chat post-processing: reasoning first, then tool calls.
    Returns (content, reasoning_content, tool_calls)."""
    content, reasoning = parse_reasoning(text)
    content, calls = parse_tool_calls(content)
    return content.strip() if calls or reasoning else content, \
        reasoning, calls
