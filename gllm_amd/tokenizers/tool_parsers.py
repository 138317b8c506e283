"""Tool-call markup parsers (reference: tokenizers/tool_parsers.py).

Turns model-emitted tool-call markup into structured OpenAI tool_calls.
Round-1 coverage: the Qwen/Hermes ``<tool_call>{json}</tool_call>``
format (Qwen2.5/Qwen3 chat templates) and a Mistral/generic
``[TOOL_CALLS]`` JSON-array fallback. Streaming variants and the
DeepSeek/Kimi DSML formats follow in a later pass.
"""

import json
import re
from typing import List, Tuple

from gllm_amd.entrypoints.protocol import FunctionCall, ToolCall

_QWEN_RE = re.compile(r"<tool_call>\s*(\{.*?\})\s*</tool_call>", re.DOTALL)
_MISTRAL_RE = re.compile(r"\[TOOL_CALLS\]\s*(\[.*\])", re.DOTALL)


def _mk_call(name: str, arguments) -> ToolCall:
    if not isinstance(arguments, str):
        arguments = json.dumps(arguments, ensure_ascii=False)
    return ToolCall(function=FunctionCall(name=name, arguments=arguments))


def parse_tool_calls(text: str, model_name: str = ""
                     ) -> Tuple[str, List[ToolCall]]:
    """Returns (content_without_markup, tool_calls)."""
    calls: List[ToolCall] = []
    matches = list(_QWEN_RE.finditer(text))
    if matches:
        for m in matches:
            try:
                obj = json.loads(m.group(1))
                calls.append(_mk_call(obj.get("name", ""),
                                      obj.get("arguments", {})))
            except json.JSONDecodeError:
                continue
        content = _QWEN_RE.sub("", text).strip()
        return content, calls
    m = _MISTRAL_RE.search(text)
    if m:
        try:
            arr = json.loads(m.group(1))
            for obj in arr:
                calls.append(_mk_call(obj.get("name", ""),
                                      obj.get("arguments", {})))
            content = _MISTRAL_RE.sub("", text).strip()
            return content, calls
        except json.JSONDecodeError:
            pass
    return text, []
