"""Request parsing for the gateway (reference: internal/apiutils/).

- SplitModelAdapter: "model_adapter" -> (model, adapter) on the first "_"
  (model names cannot contain '_'; adapter names cannot either —
  apiutils/model.go:22-30)
- Prefix extraction for PrefixHash routing: first N characters of the first
  USER message (chat) or of the prompt (completions), rune-safe
  (api/openai/v1/chat_completions.go:525-543, completions.go:134-136)
"""
from __future__ import annotations

import dataclasses
from typing import Any, Optional

from .crd import LoadBalancingSpec, Model


class APIError(Exception):
    def __init__(self, status: int, message: str):
        super().__init__(message)
        self.status = status
        self.message = message


def split_model_adapter(value: str) -> tuple[str, str]:
    model, _, adapter = value.partition("_")
    return model, adapter


def first_n_chars(s: str, n: int) -> str:
    # python strings are sequences of code points: rune-safe by construction
    return s[:n]


def extract_prefix(body: dict[str, Any], path: str, n: int) -> Optional[str]:
    if "chat/completions" in path:
        for msg in body.get("messages", []) or []:
            if msg.get("role") == "user":
                content = msg.get("content")
                if isinstance(content, list):
                    content = " ".join(
                        p.get("text", "") for p in content if isinstance(p, dict)
                    )
                return first_n_chars(content or "", n)
        return None
    prompt = body.get("prompt")
    if isinstance(prompt, list):
        prompt = prompt[0] if prompt else None
    if isinstance(prompt, str):
        return first_n_chars(prompt, n)
    return None


@dataclasses.dataclass
class ParsedRequest:
    model: str
    adapter: str
    prefix: Optional[str]
    body: dict[str, Any]
    path: str
    lb: LoadBalancingSpec


def parse_request(
    body: dict[str, Any], path: str, lookup_model, selectors: Optional[dict[str, str]] = None
) -> ParsedRequest:
    """reference: apiutils/request.go:64-232 (JSON branch).

    lookup_model(name, selectors) -> Model | None; raises APIError 404 if
    missing, 400 on missing model field, validates adapter existence.
    """
    model_field = body.get("model")
    if not model_field or not isinstance(model_field, str):
        raise APIError(400, "missing or invalid 'model' field")
    model_name, adapter = split_model_adapter(model_field)
    model: Optional[Model] = lookup_model(model_name, selectors)
    if model is None:
        raise APIError(404, f"model not found: {model_name}")
    if adapter and adapter not in {a.name for a in model.spec.adapters}:
        raise APIError(404, f"adapter not found: {adapter}")
    lb = model.spec.load_balancing
    prefix = None
    if lb.strategy == "PrefixHash":
        prefix = extract_prefix(body, path, lb.prefix_hash.prefix_char_length)
    # rewrite body for the engine: adapter name goes into the model field
    # (engine selects the LoRA adapter by model name — vLLM semantics,
    # apiutils/request.go:192-201)
    new_body = dict(body)
    new_body["model"] = adapter if adapter else model_name
    return ParsedRequest(
        model=model_name, adapter=adapter, prefix=prefix, body=new_body, path=path, lb=lb
    )
