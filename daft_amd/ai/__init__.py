"""AI provider registry + protocols (ref: /root/reference/daft/ai/ —
provider.py, protocols.py:14-50).

Providers expose TextEmbedder / ImageEmbedder / TextClassifier protocols.
Built-in providers:
  * "hash"         — deterministic feature-hash embedder (offline, no
                     weights; useful for pipelines/tests)
  * "torch"        — random-init torch models on the GPU (MFMA-backed GEMMs
                     through rocBLAS); the bench path for embedding
                     throughput without network access to checkpoints
  * "transformers" — local HuggingFace checkpoints (path must exist; no
                     downloads in this environment)
"""
from __future__ import annotations

import threading
from typing import Any, Dict, Optional, Protocol, Sequence

import numpy as np
import torch


class TextEmbedder(Protocol):
    dimensions: int

    def embed_text(self, texts: Sequence[str]) -> torch.Tensor: ...


class ImageEmbedder(Protocol):
    dimensions: int

    def embed_image(self, images: torch.Tensor) -> torch.Tensor: ...


class TextClassifier(Protocol):
    def classify_text(self, texts: Sequence[str],
                      labels: Sequence[str]) -> list: ...


class HashTextEmbedder:
    """Deterministic bag-of-words feature hashing (offline)."""

    def __init__(self, dimensions: int = 256):
        self.dimensions = dimensions

    def embed_text(self, texts):
        out = np.zeros((len(texts), self.dimensions), dtype=np.float32)
        for i, t in enumerate(texts):
            if t is None:
                continue
            for w in t.split():
                h = 2166136261
                for ch in w.encode():
                    h = ((h ^ ch) * 16777619) & 0xFFFFFFFF
                out[i, h % self.dimensions] += 1.0
            n = np.linalg.norm(out[i])
            if n > 0:
                out[i] /= n
        return torch.from_numpy(out)


class TorchImageEmbedder:
    """Random-init conv stem + MLP head in bf16 on the GPU — exercises the
    MFMA matmul path for embedding throughput benchmarks."""

    def __init__(self, dimensions: int = 512, image_size: int = 224,
                 device: Optional[str] = None, seed: int = 0):
        self.dimensions = dimensions
        if device is None:
            from ..context import get_context
            device = str(get_context().device())
        self.device = device
        dt = torch.bfloat16 if str(self.device).startswith("cuda") \
            else torch.float32
        torch.manual_seed(seed)
        self.model = torch.nn.Sequential(
            torch.nn.Conv2d(3, 64, 7, stride=4, padding=3),
            torch.nn.ReLU(),
            torch.nn.Conv2d(64, 128, 3, stride=2, padding=1),
            torch.nn.ReLU(),
            torch.nn.Conv2d(128, 256, 3, stride=2, padding=1),
            torch.nn.ReLU(),
            torch.nn.AdaptiveAvgPool2d(1),
            torch.nn.Flatten(),
            torch.nn.Linear(256, dimensions),
        ).to(self.device).to(dt).eval()
        self.dtype = dt

    @torch.no_grad()
    def embed_image(self, images: torch.Tensor) -> torch.Tensor:
        x = images.to(self.device).to(self.dtype)
        return self.model(x).to(torch.float32)


class TransformersTextEmbedder:
    def __init__(self, model_path: str, device: Optional[str] = None):
        from transformers import AutoModel, AutoTokenizer
        if device is None:
            from ..context import get_context
            device = str(get_context().device())
        self.device = device
        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.model = AutoModel.from_pretrained(model_path) \
            .to(self.device).eval()
        self.dimensions = self.model.config.hidden_size

    @torch.no_grad()
    def embed_text(self, texts):
        enc = self.tokenizer(list(texts), padding=True, truncation=True,
                             return_tensors="pt").to(self.device)
        out = self.model(**enc).last_hidden_state.mean(dim=1)
        return out.to(torch.float32).cpu()


_lock = threading.Lock()
_providers: Dict[str, Any] = {}


def load_provider(name: str = "hash", **kwargs):
    key = f"{name}:{sorted(kwargs.items())!r}"
    with _lock:
        if key in _providers:
            return _providers[key]
        if name == "hash":
            p = HashTextEmbedder(**kwargs)
        elif name == "torch":
            p = TorchImageEmbedder(**kwargs)
        elif name == "transformers":
            p = TransformersTextEmbedder(**kwargs)
        else:
            raise ValueError(f"unknown AI provider {name!r} "
                             "(available: hash, torch, transformers)")
        _providers[key] = p
        return p


def register_provider(name: str, provider) -> None:
    with _lock:
        _providers[f"{name}:[]"] = provider


# reference-name surface (daft.ai): Embedding is the vector value type
# (here a fixed-size-list / numpy vector); Provider the plugin base
from typing import Any, List, Union  # noqa: E402

Embedding = Union[List[float], "np.ndarray"]  # type: ignore[name-defined]


class Provider:
    """Base class for AI providers (ref: daft/ai Provider protocol):
    subclasses expose get_text_embedder / get_image_embedder /
    get_text_classifier and register via register_provider()."""

    name: str = "provider"

    def get_text_embedder(self, model: str = "", **kwargs):
        raise NotImplementedError

    def get_image_embedder(self, model: str = "", **kwargs):
        raise NotImplementedError

    def get_text_classifier(self, model: str = "", **kwargs):
        raise NotImplementedError
