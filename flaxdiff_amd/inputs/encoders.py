"""Conditioning encoders (reference: /root/reference/flaxdiff/inputs/encoders.py:8-98).

CLIPTextEncoder loads a HF transformers CLIP text model on ROCm (frozen; not
a perf path). There is no network in the build/bench environment, so when the
model files are unavailable the DummyTextEncoder provides a deterministic
fixed-dimension embedding for tests and synthetic benchmarks.
"""
from __future__ import annotations

from typing import List, Optional

import torch


class ConditioningEncoder:
    """tokenizer+model pair; __call__ = tokenize -> encode (reference :8-45)."""

    key: str = "text"
    model = None
    tokenizer = None

    def __call__(self, data):
        tokens = self.tokenize(data)
        return self.encode_from_tokens(tokens)

    def encode_from_tokens(self, tokens):
        raise NotImplementedError

    def tokenize(self, data):
        raise NotImplementedError

    def serialize(self) -> dict:
        raise NotImplementedError

    @staticmethod
    def deserialize(data: dict) -> "ConditioningEncoder":
        cls = ENCODER_REGISTRY[data["type"]]
        return cls.from_config(data)


class DummyTextEncoder(ConditioningEncoder):
    """Deterministic hash-based text embedding, CLIP-L/14-shaped (77, 768).

    Stands in for CLIP where model weights are unavailable (offline env);
    same tensor contract as CLIPTextEncoder.
    """

    key = "text"

    def __init__(self, seq_len: int = 77, dim: int = 768):
        self.seq_len = seq_len
        self.dim = dim

    def tokenize(self, data: List[str]):
        out = torch.zeros(len(data), self.seq_len, dtype=torch.long)
        for i, s in enumerate(data):
            toks = [hash(w) % 49408 for w in str(s).split()][: self.seq_len - 2]
            out[i, 0] = 49406
            for j, t in enumerate(toks):
                out[i, j + 1] = t
            out[i, len(toks) + 1] = 49407
        return {"input_ids": out, "attention_mask": (out != 0).long()}

    def encode_from_tokens(self, tokens):
        ids = tokens["input_ids"] if isinstance(tokens, dict) else tokens
        g = torch.Generator().manual_seed(1234)
        table = torch.randn(49408, self.dim, generator=g) * 0.02
        return table[ids]

    def serialize(self) -> dict:
        return {"type": "dummy", "seq_len": self.seq_len, "dim": self.dim}

    @classmethod
    def from_config(cls, data: dict) -> "DummyTextEncoder":
        return cls(data.get("seq_len", 77), data.get("dim", 768))


class CLIPTextEncoder(ConditioningEncoder):
    """Frozen HF CLIP text encoder; returns last_hidden_state (reference :55-98)."""

    key = "text"

    def __init__(self, modelname: str = "openai/clip-vit-large-patch14",
                 device: Optional[str] = None):
        self.modelname = modelname
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        from transformers import AutoTokenizer, CLIPTextModel
        self.tokenizer = AutoTokenizer.from_pretrained(modelname)
        self.model = CLIPTextModel.from_pretrained(modelname).to(self.device).eval()
        for p in self.model.parameters():
            p.requires_grad_(False)

    @classmethod
    def from_modelname(cls, modelname: str = "openai/clip-vit-large-patch14"):
        return cls(modelname=modelname)

    def tokenize(self, data: List[str]):
        return self.tokenizer(data, padding="max_length", max_length=77,
                              truncation=True, return_tensors="pt")

    @torch.no_grad()
    def encode_from_tokens(self, tokens):
        # transformers tokenizers return a BatchEncoding (a UserDict, NOT a
        # dict subclass) — treat any mapping with input_ids as the dict form
        if hasattr(tokens, "keys") and "input_ids" in tokens:
            ids = torch.as_tensor(tokens["input_ids"]).to(self.device)
            mask = tokens.get("attention_mask")
            if mask is not None:
                mask = torch.as_tensor(mask).to(self.device)
            out = self.model(input_ids=ids, attention_mask=mask)
        else:
            out = self.model(input_ids=torch.as_tensor(tokens).to(self.device))
        return out.last_hidden_state

    def serialize(self) -> dict:
        return {"type": "clip", "modelname": self.modelname}

    @classmethod
    def from_config(cls, data: dict) -> "CLIPTextEncoder":
        return cls(modelname=data.get("modelname", "openai/clip-vit-large-patch14"))


ENCODER_REGISTRY = {
    "clip": CLIPTextEncoder,
    "dummy": DummyTextEncoder,
}


def get_text_encoder(prefer_clip: bool = True, modelname: str = "openai/clip-vit-large-patch14"):
    """CLIP when its weights are reachable, dummy otherwise."""
    if prefer_clip:
        try:
            return CLIPTextEncoder(modelname)
        except Exception:
            pass
    return DummyTextEncoder()
