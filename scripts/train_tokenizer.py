#!/usr/bin/env python3
"""Train the offline byte-level BPE tokenizer (vocab 32k).

There is no network access for published vocabularies, so the vocabulary is
trained in-image on permissively-available English/code text: the Python
3.10 stdlib sources, this repo's own docs/sources, and the local guides.
Byte-level BPE round-trips ANY text exactly (byte fallback), which the
2000-char~=570-token rule-budget contract (convertToLLMMessageService.ts:46-48,835)
and textual-gradient critiques (apoService.ts:918-962) rely on.

Deterministic: fixed file order, fixed trainer config -> identical
tokenizer.json for every rebuild (ids must match across RCCL ranks).
"""
import glob
import os
import sys

from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

OUT = os.path.join(os.path.dirname(__file__), "..", "senweaver_amd", "engine",
                   "assets", "tokenizer.json")

def corpus_files():
    files = []
    files += sorted(glob.glob("/usr/lib/python3.10/**/*.py", recursive=True))
    files += sorted(glob.glob("/opt/skills/guides/*.md"))
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    files += sorted(glob.glob(os.path.join(repo, "*.md")))
    files += sorted(glob.glob(os.path.join(repo, "senweaver_amd", "**", "*.py"),
                             recursive=True))
    return [f for f in files if os.path.isfile(f) and os.path.getsize(f) < 2_000_000]

def iter_texts(files):
    for f in files:
        try:
            with open(f, "r", encoding="utf-8", errors="ignore") as fh:
                yield fh.read()
        except OSError:
            pass

def main():
    tok = Tokenizer(models.BPE())
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=32000 - 16,  # ids shift up by 16 to reserve specials
        min_frequency=2,
        show_progress=False,
        special_tokens=[],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    files = corpus_files()
    print(f"training on {len(files)} files", file=sys.stderr)
    tok.train_from_iterator(iter_texts(files), trainer=trainer)
    os.makedirs(os.path.dirname(OUT), exist_ok=True)
    tok.save(os.path.abspath(OUT))
    print(f"saved {OUT} vocab={tok.get_vocab_size()}", file=sys.stderr)

if __name__ == "__main__":
    main()
