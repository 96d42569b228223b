"""Tool-call + reasoning output parsers (reference postprocessor parity)."""
import json

from dynamo_amd.frontend.parsers import (parse_reasoning, parse_tool_calls,
                                         postprocess_chat)


def test_hermes_tool_tag():
    text = ('I will check the weather. <tool_call>\n'
            '{"name": "get_weather", "arguments": {"city": "Paris"}}\n'
            '</tool_call> done.')
    rest, calls = parse_tool_calls(text)
    assert len(calls) == 1
    c = calls[0]
    assert c["type"] == "function"
    assert c["function"]["name"] == "get_weather"
    assert json.loads(c["function"]["arguments"]) == {"city": "Paris"}
    assert "tool_call" not in rest and "get_weather" not in rest


def test_bare_json_tool_call():
    rest, calls = parse_tool_calls(
        ' {"name": "search", "arguments": {"q": "MI355X"}} ')
    assert len(calls) == 1 and calls[0]["function"]["name"] == "search"
    assert rest == ""


def test_multiple_and_malformed_tool_calls():
    text = ('<tool_call>{"name": "a", "arguments": {}}</tool_call>'
            '<tool_call>not json</tool_call>'
            '<tool_call>{"name": "b", "arguments": {"x": 1}}</tool_call>')
    rest, calls = parse_tool_calls(text)
    assert [c["function"]["name"] for c in calls] == ["a", "b"]
    assert "not json" in rest          # malformed tag left untouched


def test_plain_text_untouched():
    rest, calls = parse_tool_calls("just an answer, no tools")
    assert calls == [] and rest == "just an answer, no tools"
    content, reasoning = parse_reasoning("plain")
    assert content == "plain" and reasoning is None


def test_reasoning_tags():
    content, reasoning = parse_reasoning(
        "<think>step 1\nstep 2</think>The answer is 42.")
    assert content == "The answer is 42."
    assert reasoning == "step 1\nstep 2"
    # unterminated think: trailing text counts as reasoning
    content, reasoning = parse_reasoning("prefix <think>still going")
    assert content == "prefix " and reasoning == "still going"


def test_postprocess_combined():
    text = ("<think>need the weather tool</think>"
            'Sure. <tool_call>{"name": "w", "arguments": {}}</tool_call>')
    content, reasoning, calls = postprocess_chat(text)
    assert reasoning == "need the weather tool"
    assert len(calls) == 1 and calls[0]["function"]["name"] == "w"
    assert content == "Sure."


def test_incremental_detok_property():
    """Property: concatenated decode_incremental chunks == full decode,
    for any token stream and any chunking (streaming-text correctness)."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from dynamo_amd.frontend.tokenizer import ByteTokenizer

    tok = ByteTokenizer(512)

    @settings(max_examples=60, deadline=None)
    @given(st.lists(st.integers(0, 255), min_size=0, max_size=80),
           st.lists(st.integers(1, 7), min_size=1, max_size=30))
    def run(ids, cuts):
        acc = ""
        pos = 0
        i = 0
        while pos < len(ids):
            step = cuts[i % len(cuts)]
            i += 1
            nxt = min(len(ids), pos + step)
            acc += tok.decode_incremental(ids[:nxt], pos)
            pos = nxt
        assert acc == tok.decode(ids)

    run()
