"""Tokenizer wrapper: HF tokenizer from a local path, or a deterministic
synthetic tokenizer when no checkpoint is available (offline benches/tests)."""

from __future__ import annotations

from typing import List, Optional

from ..utils.logging_config import get_logger

logger = get_logger("server.tokenizer")


class SyntheticTokenizer:
    """Whitespace/byte tokenizer over a fixed vocab — lets the OpenAI API run
    end-to-end with random-weight models (no network for real tokenizers)."""

    def __init__(self, vocab_size: int = 32000):
        self.vocab_size = vocab_size
        self.eos_token_id = 2
        self.bos_token_id = 1

    def encode(self, text: str) -> List[int]:
        ids = [self.bos_token_id]
        for b in text.encode("utf-8"):
            ids.append(3 + (b % (self.vocab_size - 3)))
        return ids

    def decode(self, ids: List[int], **kw) -> str:
        return " ".join(f"<{i}>" for i in ids)

    def apply_chat_template(self, messages, add_generation_prompt=True, **kw) -> str:
        parts = [f"{m['role']}: {m['content']}" for m in messages]
        if add_generation_prompt:
            parts.append("assistant:")
        return "\n".join(parts)


class TokenizerWrapper:
    def __init__(self, model_path: Optional[str] = None, vocab_size: int = 32000):
        self.hf = None
        if model_path:
            try:
                from transformers import AutoTokenizer

                self.hf = AutoTokenizer.from_pretrained(model_path)
                logger.info("loaded HF tokenizer from %s", model_path)
            except Exception as e:  # pragma: no cover
                logger.warning("HF tokenizer load failed (%s); synthetic fallback", e)
        self.synthetic = SyntheticTokenizer(vocab_size)

    @property
    def eos_token_id(self) -> Optional[int]:
        if self.hf is not None:
            return self.hf.eos_token_id
        return self.synthetic.eos_token_id

    def encode(self, text: str) -> List[int]:
        if self.hf is not None:
            return self.hf.encode(text)
        return self.synthetic.encode(text)

    def decode(self, ids: List[int]) -> str:
        if self.hf is not None:
            return self.hf.decode(ids, skip_special_tokens=True)
        return self.synthetic.decode(ids)

    def vocab_strings(self) -> List[str]:
        """id -> token text table for constrained decoding. HF: per-id decode
        (one-time cost); synthetic: byte tokens map back to their character."""
        if self.hf is not None:
            size = len(self.hf)
            return [self.hf.decode([i]) for i in range(size)]
        v = self.synthetic
        out = [""] * v.vocab_size
        for b in range(256):
            tid = 3 + (b % (v.vocab_size - 3))
            if 0 <= tid < v.vocab_size and not out[tid]:
                out[tid] = chr(b)
        return out

    def chat_prompt_ids(self, messages: List[dict]) -> List[int]:
        if self.hf is not None and getattr(self.hf, "chat_template", None):
            return self.hf.apply_chat_template(
                messages, add_generation_prompt=True, tokenize=True
            )
        text = self.synthetic.apply_chat_template(messages)
        return self.encode(text)


class IncrementalDetokenizer:
    """Streaming detokenization with a sliding window (the standard
    incremental scheme: byte-level BPE merges across token boundaries, so
    emit only text that can no longer change; a trailing U+FFFD marks an
    incomplete multibyte sequence that must wait for more tokens).

    Replaces full-sequence re-decoding in the SSE path — that was O(n^2)
    characters per stream over a generation."""

    def __init__(self, tok: "TokenizerWrapper"):
        self.tok = tok
        self.ids: List[int] = []
        self.prefix_offset = 0
        self.read_offset = 0

    def push(self, new_ids: List[int]) -> str:
        """Append tokens; return the newly-stable text delta ('' if the tail
        is still ambiguous)."""
        self.ids.extend(new_ids)
        prefix_text = self.tok.decode(self.ids[self.prefix_offset:self.read_offset])
        new_text = self.tok.decode(self.ids[self.prefix_offset:])
        if new_text.endswith("�"):
            return ""  # incomplete utf-8 sequence: wait for the next token
        delta = new_text[len(prefix_text):]
        self.prefix_offset = self.read_offset
        self.read_offset = len(self.ids)
        return delta

    @property
    def text_so_far(self) -> str:
        """Full decode (used for final bookkeeping, not per token)."""
        return self.tok.decode(self.ids)
