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

    def decode_stream(self, ids: list[int], final: bool) -> str:
        """Streaming-safe decode: while not ``final``, a trailing
        INCOMPLETE multi-byte UTF-8 sequence is held back instead of
        decoding to U+FFFD — otherwise an already-emitted delta would
        retroactively change once the remaining continuation bytes
        arrive, and the concatenated stream would diverge from the
        final text."""
        data = bytes(i - BYTE_OFFSET for i in ids
                     if BYTE_OFFSET <= i < BYTE_OFFSET + 256)
        if not final and data:
            # find a lead byte within the last 3 positions; if its
            # sequence is unfinished, strip it
            for k in range(1, min(4, len(data) + 1)):
                b = data[-k]
                if (b & 0xC0) == 0x80:       # continuation, keep looking
                    continue
                if b >= 0xC0:                # lead byte of 2-4 byte seq
                    need = 2 if b < 0xE0 else 3 if b < 0xF0 else 4
                    if k < need:
                        data = data[:-k]
                break                        # ASCII or complete: stop
        return data.decode("utf-8", errors="replace")
