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
