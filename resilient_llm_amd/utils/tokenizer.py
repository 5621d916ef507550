"""Byte-level tokenizer for synthetic serving workloads.

There is no network for real tokenizer downloads (BASELINE.json: synthetic
prompts, random-init weights), so the serving path uses a reversible
byte-level scheme: token = byte value + 3, with PAD/BOS/EOS reserved.
Any model vocab >= 259 works (Llama-3 vocab 128256 leaves the rest of the
id space to random sampling — decoded with UTF-8 replacement so streaming
text is always valid).
"""

from __future__ import annotations

PAD_ID = 0
BOS_ID = 1
EOS_ID = 2
BYTE_OFFSET = 3


class ByteTokenizer:
    def __init__(self, vocab_size: int) -> None:
        assert vocab_size >= 259, "vocab must cover bytes + specials"
        self.vocab_size = vocab_size
        self.eos_id = EOS_ID
        self.bos_id = BOS_ID

    def encode(self, text: str, add_bos: bool = True) -> list[int]:
        ids = [BOS_ID] if add_bos else []
        ids.extend(b + BYTE_OFFSET for b in text.encode("utf-8"))
        return ids

    def decode(self, ids: list[int]) -> str:
        data = bytes(i - BYTE_OFFSET for i in ids
                     if BYTE_OFFSET <= i < BYTE_OFFSET + 256)
        return data.decode("utf-8", errors="replace")
