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


class HFTokenizer:
    """Real tokenizer loaded from a weights directory (tokenizer.json /
    tokenizer.model next to the safetensors shards) — VERDICT r01 #6:
    real-checkpoint serving needs real token boundaries and the model's
    actual EOS ids, not the byte codec's hardcoded 2."""

    def __init__(self, path: str) -> None:
        from transformers import AutoTokenizer
        self._tok = AutoTokenizer.from_pretrained(path)
        self.vocab_size = len(self._tok)
        eos = self._tok.eos_token_id
        ids = set()
        if eos is not None:
            ids.add(int(eos))
        # Llama-3 instruct checkpoints end turns with <|eot_id|> while
        # eos_token is <|end_of_text|>: both must stop generation
        for special in ("<|eot_id|>", "<|end_of_text|>", "</s>", "<|im_end|>"):
            try:
                i = self._tok.convert_tokens_to_ids(special)
            except Exception:
                i = None
            if i is not None and i >= 0 and i != getattr(
                    self._tok, "unk_token_id", None):
                ids.add(int(i))
        self.eos_ids = frozenset(ids) or frozenset({EOS_ID})
        self.eos_id = min(self.eos_ids)
        self.bos_id = self._tok.bos_token_id

    def encode(self, text: str, add_bos: bool = True) -> list:
        return self._tok.encode(text, add_special_tokens=add_bos)

    def decode(self, ids: list) -> str:
        return self._tok.decode(ids, skip_special_tokens=True)

    def decode_stream(self, ids: list, final: bool) -> str:
        text = self._tok.decode(ids, skip_special_tokens=True)
        # hold back a trailing replacement char: a multi-byte character
        # split across BPE tokens must not emit U+FFFD mid-stream
        if not final:
            text = text.rstrip("�")
        return text


def _has_tokenizer_files(path: str) -> bool:
    import os
    return any(os.path.exists(os.path.join(path, f))
               for f in ("tokenizer.json", "tokenizer.model",
                         "tokenizer_config.json"))


def load_tokenizer(weights_dir, vocab_size: int):
    """HF tokenizer when the weights dir ships one (and transformers is
    importable), else the byte codec — the synthetic-workload default."""
    if weights_dir and _has_tokenizer_files(weights_dir):
        try:
            return HFTokenizer(weights_dir)
        except Exception as e:          # missing transformers / bad files
            from .logging import log_with_timestamp
            log_with_timestamp(f"tokenizer load from {weights_dir} failed "
                               f"({e}); using byte codec", "yellow")
    return ByteTokenizer(vocab_size)
