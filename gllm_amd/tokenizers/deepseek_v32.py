"""DeepSeek-V3.2 chat encoding via the checkpoint's bundled encoder.

Parity target: reference tokenizers/deepseek_v32.py. V3.2 ships no
usable Jinja chat_template; instead the checkpoint bundles its official
DSML message encoder at ``<model_path>/encoding/encoding_dsv32.py``
(the file vLLM vendors). We import that file at runtime — it always
tracks whatever the checkpoint ships — and adapt the OpenAI-style call
surface to it. When absent, callers fall back to
``apply_chat_template``.
"""

import importlib.util
import json
import os
from typing import Any, Dict, List, Optional

_ENCODER_CACHE: Dict[str, Optional[Any]] = {}


def load_dsv32_encoder(model_path: str) -> Optional[Any]:
    """Import ``<model_path>/encoding/encoding_dsv32.py`` (cached).
    Returns the module (must expose ``encode_messages``) or None."""
    if model_path in _ENCODER_CACHE:
        return _ENCODER_CACHE[model_path]
    enc_path = os.path.join(model_path, "encoding", "encoding_dsv32.py")
    module: Optional[Any] = None
    if os.path.isfile(enc_path):
        try:
            spec = importlib.util.spec_from_file_location(
                "gllm_amd_dsv32_encoding", enc_path)
            module = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(module)
            if not hasattr(module, "encode_messages"):
                module = None
        except Exception:
            module = None
    _ENCODER_CACHE[model_path] = module
    return module


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
    norm = []
    for m in messages:
        if hasattr(m, "model_dump"):
            norm.append(m.model_dump(mode="json", exclude_none=True))
        else:
            norm.append(json.loads(json.dumps(m, default=list)))
    messages = norm
    if tools:
        messages.insert(0, {"role": "system", "tools": tools})
    drop_thinking = bool(messages) and messages[-1].get("role") == "user"
    prompt = encoder.encode_messages(
        messages, thinking_mode="thinking" if thinking else "chat",
        drop_thinking=drop_thinking)
    if not tokenize:
        return prompt
    return tokenizer.encode(prompt, add_special_tokens=False)
