#!/usr/bin/env python3
"""Train the packaged SentencePiece tokenizer (fei_amd/engine/fei16k.model).

There is no network for real tokenizer files, so the shipped model is
trained OFFLINE on a corpus assembled from this repository's own source
code, tests and documentation (plus the Python stdlib's text-rich modules)
— a realistic proxy for code-assistant traffic: agent prompts are system
text + tool schemas + code + tool output. 16k BPE vocab; byte_fallback so
any byte sequence round-trips (like Llama's tokenizer).

Run: python scripts/train_tokenizer.py   (writes fei_amd/engine/fei16k.model)
"""
import io
import os
import sys

import sentencepiece as spm

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
OUT = os.path.join(REPO, "fei_amd", "engine", "fei16k")


def gather_corpus() -> str:
    chunks = []
    roots = [REPO, os.path.dirname(os.__file__)]   # repo + python stdlib
    exts = {".py", ".md", ".hip", ".h", ".cpp", ".txt", ".json", ".cfg",
            ".toml", ".yaml"}
    budget = 24 * 1024 * 1024
    total = 0
    for root in roots:
        for dirpath, dirnames, filenames in os.walk(root):
            dirnames[:] = [d for d in dirnames
                           if d not in {".git", "__pycache__", "gpurun_out",
                                        "site-packages", "test", "idlelib"}]
            for fn in sorted(filenames):
                if os.path.splitext(fn)[1] not in exts:
                    continue
                p = os.path.join(dirpath, fn)
                try:
                    with open(p, encoding="utf-8", errors="ignore") as f:
                        t = f.read()
                except OSError:
                    continue
                chunks.append(t)
                total += len(t)
                if total > budget:
                    return "\n".join(chunks)
    return "\n".join(chunks)


def main() -> int:
    corpus = gather_corpus()
    print(f"corpus: {len(corpus) / 1e6:.1f} MB")
    # chunk the corpus preserving newlines (identity normalization below
    # keeps \n/whitespace intact — NFKC would fold newlines to spaces and
    # break code round-trips)
    chunks = [corpus[i:i + 2000] for i in range(0, len(corpus), 2000)]
    spm.SentencePieceTrainer.train(
        sentence_iterator=iter(chunks),
        model_prefix=OUT,
        vocab_size=16000,
        model_type="bpe",
        byte_fallback=True,
        normalization_rule_name="identity",
        remove_extra_whitespaces=False,
        allow_whitespace_only_pieces=True,
        character_coverage=0.9995,
        bos_id=1, eos_id=2, pad_id=3, unk_id=0,
        max_sentence_length=16384,
        num_threads=os.cpu_count() or 4,
        input_sentence_size=2000000,
        shuffle_input_sentence=True,
    )
    size = os.path.getsize(OUT + ".model")
    print(f"wrote {OUT}.model ({size // 1024} KB)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
