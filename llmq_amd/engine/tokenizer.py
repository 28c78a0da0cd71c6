"""Tokenization for the engine.

Two backends (reference consumes vLLM's engine.get_tokenizer() +
apply_chat_template, vllm_worker.py:146,175-177):

- HFTokenizer: wraps ``tokenizers``/``transformers`` when the model path is
  a checkpoint directory with a tokenizer.json (chat template honoured when
  the checkpoint ships one).
- ByteTokenizer: deterministic offline fallback (UTF-8 bytes + specials) —
  used for preset/synthetic models since this environment has no network
  for downloading real tokenizers.
"""

from __future__ import annotations

import logging
from pathlib import Path
from typing import Any, Dict, List, Optional

logger = logging.getLogger(__name__)


class ByteTokenizer:
    """UTF-8 byte-level tokenizer: ids 0..255 = bytes; then specials."""

    def __init__(self, vocab_size: int, bos_token_id: int = 1, eos_token_id: int = 2):
        # Reserve the top of the id space for BOS/EOS so byte ids stay 0..255.
        self.vocab_size = max(vocab_size, 260)
        self.bos_token_id = 256
        self.eos_token_id = 257
        self.pad_token_id = 258

    def encode(self, text: str, add_bos: bool = True) -> List[int]:
        ids = list(text.encode("utf-8"))
        return ([self.bos_token_id] + ids) if add_bos else ids

    def decode(self, ids: List[int], skip_special_tokens: bool = True) -> str:
        data = bytes(i for i in ids if i < 256)
        return data.decode("utf-8", errors="replace")

    def convert_ids_to_text_incremental(self, ids: List[int]) -> str:
        return self.decode(ids)

    def apply_chat_template(self, messages: List[Dict[str, Any]]) -> str:
        parts = []
        for m in messages:
            parts.append(f"<|{m.get('role', 'user')}|>\n{m.get('content', '')}")
        parts.append("<|assistant|>\n")
        return "\n".join(parts)


class HFTokenizer:
    def __init__(self, path: Path):
        from transformers import AutoTokenizer  # noqa: PLC0415

        self.tok = AutoTokenizer.from_pretrained(str(path), local_files_only=True)
        self.vocab_size = len(self.tok)
        self.bos_token_id = self.tok.bos_token_id
        self.eos_token_id = self.tok.eos_token_id
        self.pad_token_id = self.tok.pad_token_id

    def encode(self, text: str, add_bos: bool = True) -> List[int]:
        return self.tok.encode(text, add_special_tokens=add_bos)

    def decode(self, ids: List[int], skip_special_tokens: bool = True) -> str:
        return self.tok.decode(ids, skip_special_tokens=skip_special_tokens)

    def convert_ids_to_text_incremental(self, ids: List[int]) -> str:
        return self.decode(ids)

    def apply_chat_template(self, messages: List[Dict[str, Any]]) -> str:
        try:
            return self.tok.apply_chat_template(
                conversation=messages, tokenize=False, add_generation_prompt=True
            )
        except Exception:
            parts = [f"<|{m.get('role', 'user')}|>\n{m.get('content', '')}" for m in messages]
            parts.append("<|assistant|>\n")
            return "\n".join(parts)


def load_tokenizer(model: str, vocab_size: int, bos: int, eos: int):
    path = Path(model)
    if path.is_dir() and (
        (path / "tokenizer.json").is_file() or (path / "tokenizer.model").is_file()
    ):
        try:
            tok = HFTokenizer(path)
            # some checkpoints ship no eos/bos in tokenizer_config — fall
            # back to the model spec's ids so finish detection still works
            if tok.eos_token_id is None:
                tok.eos_token_id = eos
            if tok.bos_token_id is None:
                tok.bos_token_id = bos
            return tok
        except Exception as exc:  # pragma: no cover
            logger.warning("HF tokenizer load failed (%s); using byte tokenizer", exc)
    return ByteTokenizer(vocab_size, bos, eos)
