from gllm_amd.tokenizers.tool_parsers import parse_tool_calls


def test_qwen_tool_call():
    text = ('Let me check.\n<tool_call>\n{"name": "get_weather", '
            '"arguments": {"city": "Paris"}}\n</tool_call>')
    content, calls = parse_tool_calls(text)
    assert content == "Let me check."
    assert len(calls) == 1
    assert calls[0].function.name == "get_weather"
    assert '"city"' in calls[0].function.arguments


def test_multiple_tool_calls():
    text = ('<tool_call>{"name": "a", "arguments": {}}</tool_call>'
            '<tool_call>{"name": "b", "arguments": {"x": 1}}</tool_call>')
    content, calls = parse_tool_calls(text)
    assert [c.function.name for c in calls] == ["a", "b"]


def test_no_tool_call():
    content, calls = parse_tool_calls("just text")
    assert content == "just text" and calls == []


def test_mistral_format():
    text = '[TOOL_CALLS] [{"name": "f", "arguments": {"k": "v"}}]'
    content, calls = parse_tool_calls(text)
    assert calls and calls[0].function.name == "f"


def test_dsv32_bundled_encoder(tmp_path):
    """The checkpoint-bundled DSML encoder replaces the Jinja template
    (tokenizers/deepseek_v32.py)."""
    import os
    from gllm_amd.tokenizers.deepseek_v32 import (apply_dsv32_chat_template,
                                                  load_dsv32_encoder)
    d = tmp_path / "dsv32"
    (d / "encoding").mkdir(parents=True)
    (d / "encoding" / "encoding_dsv32.py").write_text(
        "def encode_messages(messages, thinking_mode='chat',"
        " drop_thinking=False):\n"
        "    parts = []\n"
        "    for m in messages:\n"
        "        if 'tools' in m:\n"
        "            parts.append('[tools:%d]' % len(m['tools']))\n"
        "        else:\n"
        "            parts.append('<%s>%s' % (m['role'],"
        " m.get('content','')))\n"
        "    parts.append('<mode:%s drop:%d>' % (thinking_mode,"
        " drop_thinking))\n"
        "    return ''.join(parts)\n")
    enc = load_dsv32_encoder(str(d))
    assert enc is not None

    class Tok:
        def encode(self, s, add_special_tokens=True):
            assert add_special_tokens is False
            return [ord(c) % 97 for c in s]

    msgs = [{"role": "user", "content": "hi"}]
    text = apply_dsv32_chat_template(enc, msgs, Tok(), tokenize=False)
    assert text == "<user>hi<mode:chat drop:1>"
    text2 = apply_dsv32_chat_template(enc, msgs, Tok(), tokenize=False,
                                      enable_thinking=True,
                                      tools=[{"type": "function"}])
    assert text2.startswith("[tools:1]<user>hi")
    assert "mode:thinking" in text2
    ids = apply_dsv32_chat_template(enc, msgs, Tok())
    assert isinstance(ids, list) and ids
    # missing encoder -> None (Jinja fallback)
    assert load_dsv32_encoder(str(tmp_path / "nope")) is None


def test_qwen3_xml_schema_coercion():
    from gllm_amd.tokenizers.tool_parsers import Qwen3XmlToolParser
    tools = [{"function": {
        "name": "calc",
        "parameters": {"properties": {
            "x": {"type": "integer"}, "label": {"type": "string"},
            "flags": {"type": "array"}, "on": {"type": "boolean"}}}}}]
    text = ("thinking...\n<tool_call>\n<function=calc>\n"
            "<parameter=x>\n42\n</parameter>\n"
            "<parameter=label>\n7\n</parameter>\n"
            "<parameter=flags>\n[1, 2]\n</parameter>\n"
            "<parameter=on>\ntrue\n</parameter>\n"
            "</function>\n</tool_call>")
    content, calls = Qwen3XmlToolParser().parse(text, tools)
    assert content == "thinking..."
    import json
    args = json.loads(calls[0].function.arguments)
    assert args == {"x": 42, "label": "7", "flags": [1, 2], "on": True}


def test_qwen3_xml_missing_closing_parameter():
    from gllm_amd.tokenizers.tool_parsers import Qwen3XmlToolParser
    text = ("<tool_call><function=f><parameter=a>1<parameter=b>two"
            "</function></tool_call>")
    _, calls = Qwen3XmlToolParser().parse(text, None)
    import json
    assert json.loads(calls[0].function.arguments) == {"a": "1",
                                                       "b": "two"}


def test_kimi_format():
    from gllm_amd.tokenizers.tool_parsers import KimiToolParser
    text = ("hello<|tool_calls_section_begin|>"
            "<|tool_call_begin|>functions.get_weather:0"
            "<|tool_call_argument_begin|>{\"city\": \"Paris\"}"
            "<|tool_call_end|><|tool_calls_section_end|>")
    content, calls = KimiToolParser().parse(text, None)
    assert content == "hello"
    assert calls[0].function.name == "get_weather"
    assert "Paris" in calls[0].function.arguments


def test_dsml_regex_fallback():
    from gllm_amd.tokenizers.tool_parsers import DsmlToolParser
    text = ('say<｜DSML｜function_calls>\n'
            '<｜DSML｜invoke name="get_weather">\n'
            '<｜DSML｜parameter name="city" string="true">Beijing'
            '</｜DSML｜parameter>\n'
            '<｜DSML｜parameter name="days" string="false">3'
            '</｜DSML｜parameter>\n'
            '</｜DSML｜invoke>\n</｜DSML｜function_calls>')
    content, calls = DsmlToolParser().parse(text, None)
    assert content == "say"
    import json
    args = json.loads(calls[0].function.arguments)
    assert args == {"city": "Beijing", "days": 3}


def test_streaming_parser_hermes():
    from gllm_amd.tokenizers.tool_parsers import HermesToolParser
    full = ('I will call a tool.\n<tool_call>\n{"name": "f", '
            '"arguments": {"a": 1}}\n</tool_call><tool_call>'
            '{"name": "g", "arguments": {}}</tool_call>')
    sp = HermesToolParser().stream()
    got_content = ""
    got_calls = []
    # feed in awkward chunk sizes (mid-marker splits)
    for cut in range(3, len(full) + 1, 3):
        for d in sp.feed(full[:cut]):
            if "content" in d:
                got_content += d["content"]
            else:
                got_calls.append(d["tool_call"])
    for d in sp.feed(full):
        if "content" in d:
            got_content += d["content"]
        else:
            got_calls.append(d["tool_call"])
    assert got_content == "I will call a tool.\n"
    assert [c["function"]["name"] for c in got_calls] == ["f", "g"]
    assert got_calls[0]["index"] == 0 and got_calls[1]["index"] == 1


def test_streaming_never_emits_partial_call():
    from gllm_amd.tokenizers.tool_parsers import HermesToolParser
    sp = HermesToolParser().stream()
    partial = 'ok <tool_call>{"name": "f", "argu'
    out = sp.feed(partial)
    assert all("tool_call" not in d for d in out)
    assert "".join(d.get("content", "") for d in out) == "ok "


def test_registry_selection():
    from gllm_amd.tokenizers.tool_parsers import (
        DsmlToolParser, HermesToolParser, KimiToolParser,
        MistralToolParser, Qwen3XmlToolParser, get_tool_parser)
    assert isinstance(get_tool_parser("Qwen2.5-32B"), HermesToolParser)
    assert isinstance(get_tool_parser("Qwen3.5-397B"), Qwen3XmlToolParser)
    assert isinstance(get_tool_parser("Kimi-K2.5"), KimiToolParser)
    assert isinstance(get_tool_parser("DeepSeek-V3.2-Exp"), DsmlToolParser)
    assert isinstance(get_tool_parser("Mixtral-8x7B"), MistralToolParser)
