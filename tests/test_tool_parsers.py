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
