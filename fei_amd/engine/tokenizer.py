"""Byte-level tokenizer for synthetic/random-init operation.

There is no network for real tokenizer files, so the engine ships a
self-contained byte-level tokenizer: specials + raw bytes. Token ids stay
far below any model's vocab. Round-trips arbitrary text exactly.
"""

from __future__ import annotations

from typing import List

BOS = 0
EOS = 1
PAD = 2
N_SPECIAL = 4          # 3 used + 1 reserved
BYTE_OFFSET = N_SPECIAL


class ByteTokenizer:
    vocab_size = BYTE_OFFSET + 256
    bos_id = BOS
    eos_id = EOS
    pad_id = PAD

    def encode(self, text: str, add_bos: bool = True,
               add_eos: bool = False) -> List[int]:
        ids = [BYTE_OFFSET + b for b in text.encode("utf-8")]
        if add_bos:
            ids.insert(0, BOS)
        if add_eos:
            ids.append(EOS)
        return ids

    def decode(self, ids: List[int]) -> str:
        data = bytes(i - BYTE_OFFSET for i in ids
                     if BYTE_OFFSET <= i < BYTE_OFFSET + 256)
        return data.decode("utf-8", errors="replace")


class SpmTokenizer:
    """SentencePiece tokenizer for real checkpoints (Llama-style .model
    files). The byte tokenizer remains the default — there is no network
    for tokenizer downloads — but safetensors checkpoint loading
    (models/llama.py) pairs with this when a local .model file exists.

    The same encode/decode/bos/eos surface as ByteTokenizer, so
    LocalEngine takes either interchangeably."""

    def __init__(self, model_path: str):
        import sentencepiece as spm
        self._sp = spm.SentencePieceProcessor(model_file=model_path)
        self.vocab_size = self._sp.vocab_size()
        self.bos_id = self._sp.bos_id() if self._sp.bos_id() >= 0 else 0
        self.eos_id = self._sp.eos_id() if self._sp.eos_id() >= 0 else 1
        self.pad_id = self._sp.pad_id() if self._sp.pad_id() >= 0 else \
            self.eos_id

    def encode(self, text: str, add_bos: bool = True,
               add_eos: bool = False) -> List[int]:
        ids = list(self._sp.encode(text))
        if add_bos:
            ids.insert(0, self.bos_id)
        if add_eos:
            ids.append(self.eos_id)
        return ids

    def decode(self, ids: List[int]) -> str:
        specials = {self.bos_id, self.eos_id, self.pad_id}
        # filter specials AND out-of-range ids (a random-init model samples
        # from its own, larger vocab; a real checkpoint's vocab matches)
        return self._sp.decode([i for i in ids
                                if i not in specials
                                and 0 <= i < self.vocab_size])
