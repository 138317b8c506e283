"""Incremental detokenization (reference: sequence.py:130-147).

Standard two-cursor scheme: decode from ``prefix_offset`` and emit only
the stable part, holding back text while the tail could still change
(byte-fallback / incomplete UTF-8 / merged tokens)."""

from typing import List, Optional, Tuple


class IncrementalDetokenizer:
    def __init__(self, tokenizer, skip_special_tokens: bool = True):
        self.tok = tokenizer
        self.skip_special = skip_special_tokens
        self.token_ids: List[int] = []
        self.prefix_offset = 0
        self.read_offset = 0
        self.text = ""

    def append(self, token_id: int) -> str:
        """Add one token; return newly stabilized text (may be '')."""
        self.token_ids.append(token_id)
        prefix_text = self.tok.decode(
            self.token_ids[self.prefix_offset:self.read_offset],
            skip_special_tokens=self.skip_special)
        new_text = self.tok.decode(self.token_ids[self.prefix_offset:],
                                   skip_special_tokens=self.skip_special)
        if new_text.endswith("�"):
            # incomplete byte sequence: hold back
            return ""
        delta = new_text[len(prefix_text):]
        self.prefix_offset = self.read_offset
        self.read_offset = len(self.token_ids)
        self.text += delta
        return delta


def check_stop_strings(text: str, stops: Optional[List[str]],
                       include_stop: bool = False) -> Tuple[bool, str]:
    """Return (hit, truncated_text). ``include_stop`` keeps the stop
    string in the output (OpenAI include_stop_str_in_output)."""
    if not stops:
        return False, text
    for s in stops:
        i = text.find(s)
        if i >= 0:
            return True, text[:i + len(s)] if include_stop else text[:i]
    return False, text
