"""Model registry.

The reference dispatches 38 HF Auto-class ModelTypes
(/root/reference/crates/messages/src/lib.rs:419-488 and
executors/accelerate/.../model.py). This framework implements the model
families natively (causal LM dense, causal LM MoE, GPT-2) and resolves
registry names to builders; unknown HF-only types raise with a clear message.
"""

from __future__ import annotations

from typing import Callable

import torch.nn as nn

from . import gpt2, llama, moe

_REGISTRY: dict[str, Callable[..., nn.Module]] = {}


def register(name: str, builder: Callable[..., nn.Module]) -> None:
    _REGISTRY[name] = builder


def build(name: str, **overrides) -> nn.Module:
    if name in _REGISTRY:
        return _REGISTRY[name](**overrides)
    raise KeyError(
        f"unknown model {name!r}; available: {sorted(_REGISTRY)}"
    )


def available() -> list[str]:
    return sorted(_REGISTRY)


for _name in llama.PRESETS:
    if _name == "gpt2-small":
        continue  # the true GPT-2 arch wins that name
    register(_name, (lambda n: (lambda **o: llama.build_model(n, **o)))(_name))
for _name in gpt2.PRESETS:
    register(_name, (lambda n: (lambda **o: gpt2.build_model(n))) (_name))
for _name in moe.PRESETS:
    register(_name, (lambda n: (lambda **o: moe.build_model(n, **o)))(_name))

from .llama import LlamaConfig, LlamaForCausalLM  # noqa: E402,F401
from .gpt2 import GPT2Config, GPT2ForCausalLM  # noqa: E402,F401
from .moe import MoEConfig, MoEForCausalLM  # noqa: E402,F401
