"""Streaming grammar parsers: reasoning splitter + XML tool extraction."""

from senweaver_amd.transport import (
    RawToolCall,
    ReasoningExtractor,
    XMLToolExtractor,
    ends_with_any_prefix_of,
    parse_xml_prefix_to_tool_call,
    trim_before_and_after_newlines,
)


def stream(extractor, chunks):
    """Feed cumulative strings chunk by chunk, return final state."""
    acc = ""
    for c in chunks:
        acc += c
        extractor.feed(acc)
    return extractor.finalize(acc)


def test_ends_with_any_prefix_of():
    assert ends_with_any_prefix_of("hello <thi", "<think>") == "<thi"
    assert ends_with_any_prefix_of("hello", "<think>") is None
    assert ends_with_any_prefix_of("x<think>", "<think>") == "<think>"


def test_reasoning_split_simple():
    r = ReasoningExtractor()
    text, reasoning = stream(r, ["<think>I should ", "read the file</think>", "Here is the answer"])
    assert reasoning == "I should read the file"
    assert text == "Here is the answer"


def test_reasoning_partial_tag_buffering():
    r = ReasoningExtractor()
    # the tag split across chunk boundaries must not leak into text
    text, reasoning = stream(r, ["pre <th", "ink>deep", " thought</th", "ink> post"])
    assert text == "pre  post"
    assert reasoning == "deep thought"


def test_reasoning_never_closed():
    r = ReasoningExtractor()
    text, reasoning = stream(r, ["<think>endless reasoning..."])
    assert text == ""
    assert reasoning == "endless reasoning..."


def test_reasoning_no_tags():
    r = ReasoningExtractor()
    text, reasoning = stream(r, ["plain ", "response"])
    assert text == "plain response"
    assert reasoning == ""


def test_xml_tool_streaming_partial_params():
    x = XMLToolExtractor("agent")
    acc = ""
    calls = []
    for chunk in ["Let me read it ", "<read_fi", "le><uri>/tmp/a.py</uri>",
                  "<start_line>3</start_line></read_file>"]:
        acc += chunk
        text, call = x.feed(acc)
        calls.append(call)
    text, call = x.finalize(acc)
    assert text == "Let me read it"
    assert call is not None
    assert call.name == "read_file"
    assert call.raw_params["uri"] == "/tmp/a.py"
    assert call.raw_params["start_line"] == "3"
    assert call.is_done
    assert "uri" in call.done_params
    # intermediate chunk (before close) already produced a partial call
    assert calls[2] is not None and calls[2].raw_params.get("uri") == "/tmp/a.py"


def test_xml_tool_param_aliases():
    call = parse_xml_prefix_to_tool_call(
        "read_file", "id", "<read_file><path>/x.py</path></read_file>",
        {"read_file": ["uri", "start_line", "end_line", "page_number"]})
    assert call.raw_params["uri"] == "/x.py"  # 'path' alias maps to uri


def test_xml_tool_multiline_content_trim():
    blocks = "\n<<<<<<< ORIGINAL\na\n=======\nb\n>>>>>>> UPDATED\n"
    s = f"<edit_file><uri>f.py</uri><search_replace_blocks>{blocks}</search_replace_blocks></edit_file>"
    call = parse_xml_prefix_to_tool_call(
        "edit_file", "id", s, {"edit_file": ["uri", "search_replace_blocks"]})
    # whitespace-only head/tail lines trimmed, inner newlines kept
    assert call.raw_params["search_replace_blocks"] == "<<<<<<< ORIGINAL\na\n=======\nb\n>>>>>>> UPDATED"


def test_xml_disabled_for_normal_mode():
    x = XMLToolExtractor("normal")
    text, call = x.feed("<read_file><uri>x</uri></read_file>")
    assert call is None
    assert text == "<read_file><uri>x</uri></read_file>"


def test_trim_before_after_newlines():
    assert trim_before_and_after_newlines("  \ncontent\n  ") == "content"
    assert trim_before_and_after_newlines("inline") == "inline"


def test_reasoning_chunking_invariance_property():
    """Property (hypothesis): feeding the SAME cumulative text in arbitrary
    chunk splits yields the same final (text, reasoning) as one-shot."""
    from hypothesis import given, settings, strategies as st
    from senweaver_amd.transport.grammar import ReasoningExtractor

    body = st.text(alphabet="ab<>/think ", min_size=0, max_size=60)

    @settings(derandomize=True, deadline=None)
    @given(body, st.lists(st.integers(min_value=1, max_value=10), max_size=12))
    def prop(raw, steps):
        full = f"pre {raw}<think>deep {raw}</think> post {raw}"
        one = ReasoningExtractor()
        t1, r1, _ = one.feed(full)
        t1f, r1f = one.full_text, one.full_reasoning
        inc = ReasoningExtractor()
        i = 0
        for s in steps:
            i = min(len(full), i + s)
            inc.feed(full[:i])
        inc.feed(full)
        assert inc.full_text == t1f
        assert inc.full_reasoning == r1f

    prop()


def test_xml_tool_chunking_invariance_property():
    """Property: a tool call embedded in the stream is detected with the
    same name/params regardless of chunk boundaries."""
    from hypothesis import given, settings, strategies as st
    from senweaver_amd.transport.grammar import XMLToolExtractor

    @settings(derandomize=True, deadline=None)
    @given(st.text(alphabet="abc XY\n", min_size=0, max_size=40),
           st.lists(st.integers(min_value=1, max_value=7), max_size=20))
    def prop(prefix, steps):
        full = (prefix + "<read_file><uri>some/file.txt</uri>"
                "</read_file>")
        one = XMLToolExtractor("agent")
        one.feed(full)
        ref_call = one.latest_tool_call
        assert ref_call is not None and ref_call.name == "read_file"

        inc = XMLToolExtractor("agent")
        i = 0
        for s in steps:
            i = min(len(full), i + s)
            inc.feed(full[:i])
        inc.feed(full)
        call = inc.latest_tool_call
        assert call is not None
        assert call.name == ref_call.name
        assert call.raw_params == ref_call.raw_params

    prop()
