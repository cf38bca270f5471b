"""Deterministic offline byte-level tokenizer.

There is no network to fetch a trained vocab, so the engine tokenizes at the
byte level: ids 0..2 are specials, 3..258 are raw bytes.  Model embedding
tables may be much larger (e.g. the Llama-3-8B-class generator keeps its
real 128256-entry vocab so the lm_head GEMM has the true cost); the tokenizer
simply only emits low ids.  Fully reversible, no data files.
"""

from __future__ import annotations

PAD_ID = 0
BOS_ID = 1
EOS_ID = 2
BYTE_OFFSET = 3


class ByteTokenizer:
    vocab_size = 259

    def encode(self, text: str, max_len: int | None = None, add_bos: bool = True,
               add_eos: bool = False) -> list[int]:
        ids = [BOS_ID] if add_bos else []
        ids += [b + BYTE_OFFSET for b in text.encode("utf-8")]
        if add_eos:
            ids.append(EOS_ID)
        if max_len is not None:
            ids = ids[:max_len]
        return ids

    def decode(self, ids: list[int]) -> str:
        # ids beyond the byte range can appear when a model's vocab is larger
        # than the tokenizer's (random-init logits) — drop them on decode
        data = bytes(i - BYTE_OFFSET for i in ids
                     if BYTE_OFFSET <= i < BYTE_OFFSET + 256)
        return data.decode("utf-8", errors="replace")

    def encode_batch(
        self, texts: list[str], max_len: int
    ) -> tuple[list[list[int]], list[int]]:
        """Returns (padded id lists, true lengths)."""
        seqs = [self.encode(t, max_len) for t in texts]
        lens = [len(s) for s in seqs]
        width = max(lens) if lens else 1
        padded = [s + [PAD_ID] * (width - len(s)) for s in seqs]
        return padded, lens

    def count_tokens(self, text: str) -> int:
        return 1 + len(text.encode("utf-8"))
