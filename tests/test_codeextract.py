"""extractCodeFromResult / languageHelpers rebuild (VERDICT r01 missing #6).

Includes the streaming-monotonicity property the reference promises:
"if you keep adding text, array length will strictly grow and state will
progress without going back" (extractCodeFromResult.ts).
"""

import pytest

from senweaver_amd.utils.codeextract import (
    DIVIDER, FINAL, ORIGINAL,
    extract_code_from_fim, extract_code_from_regular,
    extract_search_replace_blocks, ends_with_any_prefix_of,
    filename_to_language, markdown_language_to_id,
)


def test_code_block_stripping():
    s, _, _ = extract_code_from_regular("```python\nprint(1)\n```")
    assert s == "print(1)"
    s, _, _ = extract_code_from_regular("```\nx = 2\n```\n")
    assert s == "x = 2"
    # no fences: returned untouched
    s, _, _ = extract_code_from_regular("plain text")
    assert s == "plain text"


def test_partial_closing_fence_stripped():
    # streaming: the trailing partial ``` must not leak into the value
    # (the newline before the still-growing fence stays, as in the reference)
    s, _, _ = extract_code_from_regular("```py\ncode\n``")
    assert s == "code\n"


def test_fim_extraction():
    s, _, _ = extract_code_from_fim("```\n<MID>the middle</MID>\n```", "MID")
    assert s == "the middle"
    s, _, _ = extract_code_from_fim("<MID>partial", "MID")
    assert s == "partial"


def test_search_replace_done_block():
    text = (f"{ORIGINAL}\nold line\n{DIVIDER}\nnew line\n{FINAL}\n")
    blocks = extract_search_replace_blocks(text)
    assert len(blocks) == 1
    b = blocks[0]
    assert (b.state, b.orig, b.final) == ("done", "old line", "new line")


def test_search_replace_multiple_blocks():
    one = f"{ORIGINAL}\na\n{DIVIDER}\nb\n{FINAL}\n"
    blocks = extract_search_replace_blocks(one * 3)
    assert [b.state for b in blocks] == ["done"] * 3
    assert [b.orig for b in blocks] == ["a"] * 3


def test_streaming_monotonicity():
    """Feeding prefixes of a full stream: block count never shrinks and
    per-block state never regresses."""
    full = (f"prefix text\n{ORIGINAL}\nalpha\nbeta\n{DIVIDER}\n"
            f"gamma\n{FINAL}\n tail {ORIGINAL}\nx\n{DIVIDER}\ny\n{FINAL}\n")
    rank = {"writingOriginal": 0, "writingFinal": 1, "done": 2}
    prev_states = []
    for cut in range(len(full) + 1):
        blocks = extract_search_replace_blocks(full[:cut])
        states = [rank[b.state] for b in blocks]
        assert len(states) >= len(prev_states), cut
        for old, new in zip(prev_states, states):
            assert new >= old, cut
        prev_states = states
    assert prev_states == [2, 2]


def test_partial_divider_not_leaked_into_orig():
    text = f"{ORIGINAL}\ncontent\n===="
    blocks = extract_search_replace_blocks(text)
    assert blocks[0].state == "writingOriginal"
    assert blocks[0].orig == "content"


def test_ends_with_any_prefix_of():
    assert ends_with_any_prefix_of("abc\n===", "\n=======") == "\n==="
    assert ends_with_any_prefix_of("abc", "\n====") is None


def test_language_helpers():
    assert filename_to_language("src/foo.py") == "python"
    assert filename_to_language("Dockerfile") == "dockerfile"
    assert filename_to_language("weird.xyz") == "plaintext"
    assert markdown_language_to_id("ts") == "typescript"
    assert markdown_language_to_id("C++") == "cpp"
    assert markdown_language_to_id("") == "plaintext"
    assert markdown_language_to_id("python") == "python"


def test_search_replace_fuzz_never_crashes_and_streams_monotonic():
    """Property (hypothesis): for ANY text sliced at arbitrary chunk
    boundaries, extract_search_replace_blocks never raises, and blocks
    marked done in an earlier prefix keep identical orig/final in every
    longer prefix (the monotonicity the streaming applier relies on)."""
    from hypothesis import given, settings, strategies as st
    from senweaver_amd.utils.codeextract import extract_search_replace_blocks

    fragments = st.lists(st.sampled_from([
        "<<<<<<< ORIGINAL\n", "=======\n", ">>>>>>> UPDATED\n",
        "code line\n", "x = 1", "<<<<", ">>>", "====", "\n", "random text ",
    ]), min_size=0, max_size=14)

    @settings(max_examples=120, deadline=None)
    @given(fragments, st.integers(min_value=1, max_value=9))
    def check(frags, step):
        s = "".join(frags)
        done_seen = {}
        for cut in range(0, len(s) + 1, step):
            blocks = extract_search_replace_blocks(s[:cut])
            for i, b in enumerate(blocks):
                if b.state == "done":
                    if i in done_seen:
                        assert done_seen[i] == (b.orig, b.final)
                    else:
                        done_seen[i] = (b.orig, b.final)
        extract_search_replace_blocks(s)  # full text never raises

    check()
