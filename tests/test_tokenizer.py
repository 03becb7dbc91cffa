"""Byte-level BPE tokenizer: round-trip, chars/token contract, determinism.

The reference budgets text by CHARS_PER_TOKEN = 3.5
(convertToLLMMessageService.ts:46-48) and the 2000-char APO rule budget is
documented as ~570 tokens (:835); the in-repo tokenizer must make those
equivalences physically true, not nominal.
"""

import pytest

from senweaver_amd.engine import tokenizer as tok

PROMPT_CORPUS = [
    "You are a helpful coding assistant. Always explain your reasoning "
    "before making changes, and prefer small, reviewable edits.",
    "When the user reports a failing test, first reproduce it, then read "
    "the implicated source files before proposing a fix.",
    "- Use the read_file tool before editing any file.\n"
    "- Never run destructive commands without approval.\n"
    "- Summarize tool output instead of quoting it in full.",
    "The function should return the number of tokens consumed by the "
    "system message, including the APO optimized rules section.",
    "def compute_reward(spans):\n    total = 0.0\n    for s in spans:\n"
    "        total += s.weight * s.value\n    return total / len(spans)",
]


def test_roundtrip_exact():
    t = tok.BPETokenizer(128256)
    for s in PROMPT_CORPUS + ["unicode: 日本語 émojis 🙂 tabs\t\tnewlines\n\n"]:
        assert t.decode(t.encode(s)) == s


def test_chars_per_token_contract():
    t = tok.BPETokenizer(128256)
    text = "\n".join(PROMPT_CORPUS)
    n = len(t.encode(text))
    cpt = len(text) / n
    assert 2.5 <= cpt <= 4.5, f"chars/token {cpt} outside 3.5 +- 1"


def test_rule_budget_token_equivalence():
    # 2000 chars of realistic rule text ~= 570 tokens (within 2x band)
    rules = ("- Always check the exit code of every command you run.\n" * 40)[:2000]
    t = tok.BPETokenizer(128256)
    n = len(t.encode(rules))
    assert 285 <= n <= 1140, n


def test_determinism_and_specials():
    a = tok.BPETokenizer(128256)
    b = tok.BPETokenizer(128256)
    s = PROMPT_CORPUS[0]
    assert a.encode(s) == b.encode(s)
    # no encoded id may collide with the reserved special band
    assert min(a.encode(s)) >= tok._SPECIAL_MAX


def test_folded_mode_fits_tiny_vocab():
    t = tok.BPETokenizer(512)
    ids = t.encode("fold me into a tiny vocabulary " * 20)
    assert ids and all(256 <= i < 512 for i in ids)


def test_mixtral_vocab_is_full_mode():
    # 32000 = 16 specials + 31984 BPE ids: exactly fits
    t = tok.BPETokenizer(32000)
    s = "expert routing with grouped GEMM"
    assert t.decode(t.encode(s)) == s


def test_max_tokens_truncation():
    t = tok.BPETokenizer(128256)
    ids = t.encode("word " * 100, max_tokens=7)
    assert len(ids) == 7


def test_for_vocab_prefers_bpe():
    assert isinstance(tok.for_vocab(128256), tok.BPETokenizer)


def test_roundtrip_property():
    """Property (hypothesis): encode->decode is lossless for printable
    text on the full 32k tokenizer (byte-level BPE is invertible)."""
    from hypothesis import given, settings, strategies as st
    from senweaver_amd.engine.tokenizer import BPETokenizer

    t = BPETokenizer(vocab_size=128256)

    @settings(max_examples=60, deadline=None)
    @given(st.text(alphabet=st.characters(min_codepoint=32, max_codepoint=0x2FFF),
                   min_size=0, max_size=120))
    def check(s):
        assert t.decode(t.encode(s)) == s

    check()


def test_roundtrip_multibyte_and_newlines():
    from senweaver_amd.engine.tokenizer import BPETokenizer
    t = BPETokenizer(vocab_size=128256)
    for s in ["", "\n\n\t  mixed\nlines\n", "café über 中文 \U0001f600",
              "def f(x):\n    return x * 2  # comment\n"]:
        assert t.decode(t.encode(s)) == s
