"""Tokenizer loading with a robust fast-tokenizer fallback.

AutoTokenizer resolves the class from config.json's model_type, which
loses against a plain tokenizer.json + tokenizer_config.json pair (it
can instantiate e.g. a slow Qwen2Tokenizer with an empty vocab when
vocab.json/merges.txt are absent). If the Auto result has an empty
vocab and a tokenizer.json exists, load PreTrainedTokenizerFast
directly with the config's special tokens + chat template.
"""

import json
import os


def load_tokenizer(model_path: str):
    if not model_path:
        return None
    has_any = any(os.path.exists(os.path.join(model_path, f))
                  for f in ("tokenizer.json", "tokenizer.model",
                            "tokenizer_config.json"))
    if not has_any:
        return None
    tok = None
    try:
        from transformers import AutoTokenizer
        tok = AutoTokenizer.from_pretrained(model_path,
                                            trust_remote_code=True)
    except Exception:
        tok = None
    def _broken(t) -> bool:
        if t is None:
            return True
        try:
            return len(t.encode("probe", add_special_tokens=False)) == 0
        except Exception:
            return True

    tj = os.path.join(model_path, "tokenizer.json")
    if _broken(tok) and os.path.exists(tj):
        from transformers import PreTrainedTokenizerFast
        kwargs = {}
        tc = os.path.join(model_path, "tokenizer_config.json")
        if os.path.exists(tc):
            with open(tc) as f:
                c = json.load(f)
            for k in ("eos_token", "bos_token", "unk_token", "pad_token",
                      "chat_template", "model_max_length"):
                if c.get(k) is not None:
                    kwargs[k] = c[k]
        tok = PreTrainedTokenizerFast(tokenizer_file=tj, **kwargs)
    return tok
