"""Tool-call markup parsers + streaming (reference parity:
tokenizers/tool_parsers.py, 673 lines — Qwen/Hermes JSON, Qwen3.5 XML,
Kimi K2 sections, DeepSeek DSML, Mistral arrays, schema-aware argument
coercion, incremental stream parsing).

Formats are dictated by each model family's chat template, so the
markers below match the models' actual output grammar:

* Hermes/Qwen2.5:  ``<tool_call>{json}</tool_call>``
* Qwen3.5 XML:     ``<tool_call><function=f><parameter=k>v</parameter>
                   ...</function></tool_call>`` — parameter values are
                   raw text typed against the tool's JSON schema
* Kimi K2:         ``<|tool_calls_section_begin|>`` section of
                   ``<|tool_call_begin|>functions.f:i
                   <|tool_call_argument_begin|>{json}<|tool_call_end|>``
* DeepSeek DSML:   ``<｜DSML｜function_calls>`` blocks of ``invoke`` /
                   typed ``parameter`` tags; delegated to the
                   checkpoint's bundled decoder when available
* Mistral:         ``[TOOL_CALLS][{...}, ...]``

Streaming protocol: ``stream(tools)`` returns a parser whose
``feed(full_text)`` is called with the cumulative decoded text and
returns the list of deltas now emittable — plain content up to the
first tool marker as it grows, then one delta per COMPLETED call.
"""

import json
import re
from typing import Any, Dict, List, Optional, Tuple

from gllm_amd.entrypoints.protocol import FunctionCall, ToolCall


# ----------------------------------------------------------- schema typing
def _schema_types(schema: Any) -> List[str]:
    if not isinstance(schema, dict):
        return []
    t = schema.get("type")
    if isinstance(t, str):
        return [t]
    if isinstance(t, list):
        return [x for x in t if isinstance(x, str)]
    if "anyOf" in schema:
        out: List[str] = []
        for sub in schema["anyOf"]:
            out.extend(_schema_types(sub))
        return out
    return []


def _param_schemas(tools, func_name: str) -> Optional[Dict[str, Any]]:
    for t in tools or []:
        fn = t.get("function", t) if isinstance(t, dict) else None
        if fn and fn.get("name") == func_name:
            params = fn.get("parameters") or {}
            return params.get("properties") or {}
    return None


def _coerce(value: str, schema: Any):
    """Type a raw text value against its JSON-schema parameter types.
    String-typed params stay strings (unconditional json.loads would
    break them); numeric/bool/array/object params parse."""
    types = _schema_types(schema)
    if not types or "string" in types:
        return value
    s = value.strip()
    try:
        if "boolean" in types:
            if s.lower() in ("true", "false"):
                return s.lower() == "true"
        if "integer" in types:
            return int(s)
        if "number" in types:
            f = float(s)
            return int(f) if f.is_integer() and "." not in s else f
        if "array" in types or "object" in types:
            return json.loads(s)
        if "null" in types and s.lower() in ("null", "none", ""):
            return None
    except (ValueError, json.JSONDecodeError):
        pass
    return value


def _coerce_args(args: Dict[str, str], tools,
                 func_name: str) -> Dict[str, Any]:
    props = _param_schemas(tools, func_name)
    if props is None:
        return dict(args)
    return {k: _coerce(v, props.get(k)) for k, v in args.items()}


def _args_json(arguments) -> str:
    if isinstance(arguments, str):
        return arguments
    return json.dumps(arguments, ensure_ascii=False)


def _mk_call(name: str, arguments) -> ToolCall:
    return ToolCall(function=FunctionCall(name=name,
                                          arguments=_args_json(arguments)))


# ----------------------------------------------------------------- base
class ToolParser:
    name = "base"
    # markers whose PREFIX at the end of the text must be held back
    # from content streaming (a half-arrived "<tool_c" is not content)
    start_markers: Tuple[str, ...] = ()

    def content_head(self, text: str) -> str:
        """Text before the first tool-call marker (streamed as plain
        content)."""
        return text

    def parse(self, text: str, tools=None
              ) -> Tuple[Optional[str], List[ToolCall]]:
        raise NotImplementedError

    def stream(self, tools=None) -> "StreamingToolParser":
        return StreamingToolParser(self, tools)


class StreamingToolParser:
    """Incremental parse over the cumulative decoded text.

    ``feed`` returns the deltas now safe to emit: the growing plain
    content BEFORE the first marker, then each tool call once its
    closing marker has arrived (complete calls only — a partially
    generated call is never surfaced)."""

    def __init__(self, parser: ToolParser, tools=None):
        self.parser = parser
        self.tools = tools
        self._content_sent = 0
        self._calls_sent = 0

    @property
    def emitted_tool_calls(self) -> bool:
        return self._calls_sent > 0

    def _held(self, head: str) -> int:
        """Length of the trailing piece that may be the start of a
        marker (withheld until it resolves either way)."""
        best = 0
        for mk in self.parser.start_markers:
            for k in range(min(len(mk) - 1, len(head)), 0, -1):
                if head.endswith(mk[:k]):
                    best = max(best, k)
                    break
        return best

    def feed(self, full_text: str) -> List[dict]:
        out: List[dict] = []
        head = self.parser.content_head(full_text)
        emit_to = len(head)
        if emit_to == len(full_text):  # no marker yet: hold a prefix
            emit_to -= self._held(head)
        if self._content_sent < emit_to:
            out.append({"content": head[self._content_sent:emit_to]})
            self._content_sent = emit_to
        _, calls = self.parser.parse(full_text, self.tools)
        while self._calls_sent < len(calls):
            c = calls[self._calls_sent]
            out.append({"tool_call": {
                "index": self._calls_sent,
                "id": c.id,
                "type": "function",
                "function": {"name": c.function.name,
                             "arguments": c.function.arguments},
            }})
            self._calls_sent += 1
        return out


# ---------------------------------------------------------------- hermes
class HermesToolParser(ToolParser):
    """Qwen2/2.5 + Hermes: JSON object inside <tool_call> tags."""

    name = "hermes"
    _START = "<tool_call>"
    start_markers = ("<tool_call>",)
    _RE = re.compile(r"<tool_call>\s*(\{.*?\})\s*</tool_call>", re.DOTALL)

    def content_head(self, text: str) -> str:
        return text.split(self._START, 1)[0]

    def parse(self, text, tools=None):
        calls = []
        for m in self._RE.finditer(text):
            try:
                obj = json.loads(m.group(1))
            except json.JSONDecodeError:
                continue
            calls.append(_mk_call(obj.get("name", ""),
                                  obj.get("arguments",
                                          obj.get("parameters", {}))))
        if not calls:
            return text, []
        return self._RE.sub("", text).strip() or None, calls


# ---------------------------------------------------------------- qwen3 xml
class Qwen3XmlToolParser(ToolParser):
    """Qwen3.5 XML form; values typed against the tool schema. The
    closing </parameter> may be dropped by the model — a value ends at
    the next <parameter=, </function>, or end of body."""

    name = "qwen3_xml"
    _START = "<tool_call>"
    start_markers = ("<tool_call>",)
    _FUNC = re.compile(r"<function=([^>\n]+)>(.*?)</function>", re.DOTALL)
    _PARAM = re.compile(
        r"<parameter=([^>\n]+)>(.*?)(?:</parameter>|(?=<parameter=)|\Z)",
        re.DOTALL)

    def content_head(self, text: str) -> str:
        return text.split(self._START, 1)[0]

    def parse(self, text, tools=None):
        if self._START not in text:
            return text, []
        calls = []
        for fm in self._FUNC.finditer(text):
            name = fm.group(1).strip()
            if not name:
                continue
            raw = {k.strip(): v.strip()
                   for k, v in self._PARAM.findall(fm.group(2)) if k.strip()}
            calls.append(_mk_call(name, _coerce_args(raw, tools, name)))
        return self.content_head(text).strip() or None, calls


# ---------------------------------------------------------------- kimi
class KimiToolParser(ToolParser):
    """Kimi K2/K2.5 section form; ids look like functions.NAME:IDX."""

    name = "kimi"
    _START = "<|tool_calls_section_begin|>"
    start_markers = ("<|tool_calls_section_begin|>",)
    _CALL = re.compile(
        r"<\|tool_call_begin\|>\s*([^\s<]+?)\s*"
        r"<\|tool_call_argument_begin\|>\s*(.*?)\s*<\|tool_call_end\|>",
        re.DOTALL)

    def content_head(self, text: str) -> str:
        return text.split(self._START, 1)[0]

    def parse(self, text, tools=None):
        if self._START not in text:
            return text, []
        calls = []
        for fid, args in self._CALL.findall(text):
            name = fid.split(":", 1)[0]
            name = name[len("functions."):] if \
                name.startswith("functions.") else name
            if name:
                calls.append(_mk_call(name, args.strip()))
        return self.content_head(text).strip() or None, calls


# ---------------------------------------------------------------- dsml
class DsmlToolParser(ToolParser):
    """DeepSeek-V3.2 DSML blocks. When the checkpoint's bundled decoder
    (encoding_dsv32) is injected, parsing delegates to it so typed
    arguments match upstream exactly; otherwise a tolerant regex path
    extracts names + parameters (the ｜DSML｜ prefix is optional — some
    decodes drop the special token)."""

    name = "dsml"
    _STARTS = ("<｜DSML｜function_calls", "<function_calls")
    start_markers = _STARTS
    _INVOKE = re.compile(
        r"<(?:｜DSML｜)?invoke\s+name=\"([^\"]+)\">(.*?)"
        r"</(?:｜DSML｜)?invoke>", re.DOTALL)
    _PARAM = re.compile(
        r"<(?:｜DSML｜)?parameter\s+name=\"([^\"]+)\"\s+"
        r"string=\"(true|false)\">(.*?)</(?:｜DSML｜)?parameter>",
        re.DOTALL)

    def __init__(self, encoder=None):
        self.encoder = encoder

    def _start(self, text: str) -> int:
        idxs = [text.find(s) for s in self._STARTS]
        idxs = [i for i in idxs if i >= 0]
        return min(idxs) if idxs else -1

    def content_head(self, text: str) -> str:
        i = self._start(text)
        return text if i < 0 else text[:i]

    def _parse_bundled(self, text):
        msg = self.encoder.parse_message_from_completion_text(
            text, role="assistant")
        calls = []
        for tc in msg.get("tool_calls") or []:
            fn = tc.get("function", {})
            calls.append(_mk_call(fn.get("name", ""),
                                  fn.get("arguments", "{}")))
        return (msg.get("content") or None), calls

    def parse(self, text, tools=None):
        if self._start(text) < 0:
            return text, []
        if self.encoder is not None:
            try:
                return self._parse_bundled(text)
            except Exception:
                pass  # fall through to the regex path
        calls = []
        for name, body in self._INVOKE.findall(text):
            args: Dict[str, Any] = {}
            for key, is_str, val in self._PARAM.findall(body):
                if is_str == "true":
                    args[key] = val
                else:
                    try:
                        args[key] = json.loads(val)
                    except json.JSONDecodeError:
                        args[key] = val
            calls.append(_mk_call(name, args))
        return self.content_head(text).strip() or None, calls


# ---------------------------------------------------------------- mistral
class MistralToolParser(ToolParser):
    name = "mistral"
    _START = "[TOOL_CALLS]"
    start_markers = ("[TOOL_CALLS]",)
    _RE = re.compile(r"\[TOOL_CALLS\]\s*(\[.*\])", re.DOTALL)

    def content_head(self, text: str) -> str:
        return text.split(self._START, 1)[0]

    def parse(self, text, tools=None):
        m = self._RE.search(text)
        if not m:
            return text, []
        try:
            arr = json.loads(m.group(1))
        except json.JSONDecodeError:
            return text, []
        calls = [_mk_call(o.get("name", ""), o.get("arguments", {}))
                 for o in arr]
        return self._RE.sub("", text).strip() or None, calls


# ---------------------------------------------------------------- registry
def get_tool_parser(model_name: str = "", architecture: str = "",
                    encoder=None) -> ToolParser:
    """Pick a parser by model/architecture name. ``encoder`` is the
    DSV3.2 bundled decoder (entrypoints inject it when the checkpoint
    ships one)."""
    key = f"{model_name} {architecture}".lower()
    if "deepseekv32" in key.replace("_", "").replace("-", "").replace(".", "") or \
            "dsml" in key:
        return DsmlToolParser(encoder)
    if "kimi" in key:
        return KimiToolParser()
    if "mistral" in key or "mixtral" in key:
        return MistralToolParser()
    if "qwen3.5" in key or "qwen3_5" in key or "qwen3-next" in key:
        return Qwen3XmlToolParser()
    return HermesToolParser()


# ------------------------------------------------------- legacy facade
def parse_tool_calls(text: str, model_name: str = "", tools=None,
                     encoder=None) -> Tuple[str, List[ToolCall]]:
    """One-shot parse used by the non-stream chat path. Tries the
    model's parser first, then falls back across formats so a
    mis-labelled model still yields structured calls."""
    primary = get_tool_parser(model_name, encoder=encoder)
    content, calls = primary.parse(text, tools)
    if calls:
        return (content or ""), calls
    for cls in (HermesToolParser, Qwen3XmlToolParser, KimiToolParser,
                MistralToolParser):
        if isinstance(primary, cls):
            continue
        p = cls()
        content, calls = p.parse(text, tools)
        if calls:
            return (content or ""), calls
    dsml = DsmlToolParser(encoder)
    if not isinstance(primary, DsmlToolParser):
        content, calls = dsml.parse(text, tools)
        if calls:
            return (content or ""), calls
    return text, []
