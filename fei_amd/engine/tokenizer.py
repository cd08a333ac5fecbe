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
