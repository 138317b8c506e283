"""DeepSeek-V3.2 chat encoding via the checkpoint's bundled encoder.

Parity target: reference tokenizers/deepseek_v32.py. V3.2 ships no
usable Jinja chat_template; instead the checkpoint bundles its official
DSML message encoder at ``<model_path>/encoding/encoding_dsv32.py``
(the file vLLM vendors). We import that file at runtime — it always
tracks whatever the checkpoint ships — and adapt the OpenAI-style call
surface to it. When absent, callers fall back to
``apply_chat_template``.
"""

import json
import pathlib
from functools import lru_cache
from importlib import util as _imp
from typing import Any, List, Optional


@lru_cache(maxsize=16)
def load_dsv32_encoder(model_path: str) -> Optional[Any]:
    """Import ``<model_path>/encoding/encoding_dsv32.py``.
    Returns the module (must expose ``encode_messages``) or None."""
    src = pathlib.Path(model_path) / "encoding" / "encoding_dsv32.py"
    if not src.is_file():
        return None
    try:
        spec = _imp.spec_from_file_location("gllm_amd_dsv32_encoding",
                                            str(src))
        mod = _imp.module_from_spec(spec)
        spec.loader.exec_module(mod)
    except Exception:
        return None
    return mod if callable(getattr(mod, "encode_messages", None)) else None


def _as_plain_dict(m) -> dict:
    if hasattr(m, "model_dump"):
        return m.model_dump(mode="json", exclude_none=True)
    return json.loads(json.dumps(m, default=list))


def apply_dsv32_chat_template(encoder: Any, messages: List[dict],
                              tokenizer: Any, *,
                              tools: Optional[List[dict]] = None,
                              tokenize: bool = True, **kwargs):
    """Render messages with the official encoder.

    - ``thinking`` / ``enable_thinking`` kwargs pick thinking_mode;
    - tools ride a leading system message;
    - prior-turn reasoning is dropped when a fresh user turn arrives;
    - the encoder emits BOS itself, so tokenize with
      add_special_tokens=False.
    """
    thinking = bool(kwargs.get("thinking", False)
                    or kwargs.get("enable_thinking", False))
    convo = [_as_plain_dict(m) for m in messages]
    if tools:
        convo = [{"role": "system", "tools": tools}] + convo
    fresh_user_turn = bool(convo) and convo[-1].get("role") == "user"
    prompt = encoder.encode_messages(
        convo, thinking_mode="thinking" if thinking else "chat",
        drop_thinking=fresh_user_turn)
    if not tokenize:
        return prompt
    return tokenizer.encode(prompt, add_special_tokens=False)
