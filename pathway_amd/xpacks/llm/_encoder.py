"""MI355X-native sentence-encoder forward (bge-small-class architecture).

The reference delegates embedding to the external sentence-transformers
library (xpacks/llm/embedders.py:454-540, device + batch 1024).  Here the
encoder forward is implemented directly: a BERT-small-shape transformer
encoder (L=12, H=384, heads=12, ffn=1536 — bge-small-en-v1.5 geometry) run
in bf16 on the GPU; GEMMs hit hipBLASLt MFMA paths through torch-ROCm, and
the module is wrapped in torch.compile-free explicit code so hipGraph
capture of the fixed-shape microbatch is possible.

There is no network access for checkpoint downloads, so weights are
deterministic random-init (seeded) and tokenization is hash-based — the
geometry, FLOPs and memory traffic match the real model, which is what the
serving benchmark measures (BASELINE config 3: synthetic data, random-init
weights).
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np
import torch
import torch.nn.functional as F

from pathway_amd.internals.api import xxh64


@dataclass
class EncoderConfig:
    dim: int = 384
    layers: int = 12
    heads: int = 12
    ffn: int = 1536
    vocab: int = 30522
    max_len: int = 512
    seed: int = 1234


class NativeEncoder(torch.nn.Module):
    def __init__(self, cfg: EncoderConfig | None = None, device="cpu", dtype=None):
        super().__init__()
        self.cfg = cfg or EncoderConfig()
        c = self.cfg
        if dtype is None:
            dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
        self.dtype = dtype
        g = torch.Generator().manual_seed(c.seed)

        def p(*shape, scale=0.02):
            return torch.nn.Parameter(
                (torch.randn(*shape, generator=g) * scale).to(dtype)
            )

        self.tok_emb = p(c.vocab, c.dim)
        self.pos_emb = p(c.max_len, c.dim)
        self.blocks = torch.nn.ParameterList()
        self.layer_params = []
        for _ in range(c.layers):
            lp = {
                "qkv_w": p(c.dim, 3 * c.dim),
                "qkv_b": p(3 * c.dim, scale=0.0),
                "proj_w": p(c.dim, c.dim),
                "proj_b": p(c.dim, scale=0.0),
                "ln1_w": torch.nn.Parameter(torch.ones(c.dim, dtype=dtype)),
                "ln1_b": torch.nn.Parameter(torch.zeros(c.dim, dtype=dtype)),
                "fc1_w": p(c.dim, c.ffn),
                "fc1_b": p(c.ffn, scale=0.0),
                "fc2_w": p(c.ffn, c.dim),
                "fc2_b": p(c.dim, scale=0.0),
                "ln2_w": torch.nn.Parameter(torch.ones(c.dim, dtype=dtype)),
                "ln2_b": torch.nn.Parameter(torch.zeros(c.dim, dtype=dtype)),
            }
            self.layer_params.append(lp)
            for v in lp.values():
                self.blocks.append(v)
        self.to(device)
        self.device = torch.device(device)
        self.eval()

    @torch.no_grad()
    def forward(self, ids: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        """ids (b, s) int64, mask (b, s) bool → (b, dim) f32 normalized."""
        c = self.cfg
        b, s = ids.shape
        x = self.tok_emb[ids] + self.pos_emb[:s].unsqueeze(0)
        attn_bias = torch.where(
            mask.unsqueeze(1).unsqueeze(2),
            torch.zeros((), dtype=x.dtype, device=x.device),
            torch.full((), float("-inf"), dtype=x.dtype, device=x.device),
        )
        for lp in self.layer_params:
            h = F.layer_norm(x, (c.dim,), lp["ln1_w"], lp["ln1_b"])
            qkv = h @ lp["qkv_w"] + lp["qkv_b"]
            q, k, v = qkv.split(c.dim, dim=-1)
            hd = c.dim // c.heads
            q = q.view(b, s, c.heads, hd).transpose(1, 2)
            k = k.view(b, s, c.heads, hd).transpose(1, 2)
            v = v.view(b, s, c.heads, hd).transpose(1, 2)
            o = F.scaled_dot_product_attention(q, k, v, attn_mask=attn_bias)
            o = o.transpose(1, 2).reshape(b, s, c.dim)
            x = x + o @ lp["proj_w"] + lp["proj_b"]
            h = F.layer_norm(x, (c.dim,), lp["ln2_w"], lp["ln2_b"])
            x = x + F.gelu(h @ lp["fc1_w"] + lp["fc1_b"]) @ lp["fc2_w"] + lp["fc2_b"]
        # masked mean pool + L2 normalize (sentence-transformers convention)
        m = mask.unsqueeze(-1).to(x.dtype)
        pooled = (x * m).sum(1) / m.sum(1).clamp(min=1)
        return F.normalize(pooled.to(torch.float32), dim=-1)

    def tokenize(self, texts: list[str], max_len: int | None = None):
        c = self.cfg
        max_len = min(max_len or c.max_len, c.max_len)
        cache = getattr(self, "_tok_cache", None)
        if cache is None:
            cache = self._tok_cache = {}
        vocab_span = c.vocab - 3
        tok_rows = []
        for t in texts:
            words = (t or "").lower().split()[:max_len]
            ids = []
            for w in words:
                tid = cache.get(w)
                if tid is None:
                    tid = 3 + (xxh64(w.encode(), 77) % vocab_span)
                    if len(cache) < 1_000_000:
                        cache[w] = tid
                ids.append(tid)
            if not ids:
                ids = [1]
            tok_rows.append(ids)
        s = max(len(r) for r in tok_rows)
        ids = torch.zeros((len(texts), s), dtype=torch.int64)
        mask = torch.zeros((len(texts), s), dtype=torch.bool)
        for i, r in enumerate(tok_rows):
            ids[i, : len(r)] = torch.tensor(r, dtype=torch.int64)
            mask[i, : len(r)] = True
        return ids.to(self.device), mask.to(self.device)

    @torch.no_grad()
    def encode(self, texts: list[str], batch_size: int = 1024) -> list[np.ndarray]:
        out: list[np.ndarray] = []
        for i in range(0, len(texts), batch_size):
            chunk = texts[i : i + batch_size]
            ids, mask = self.tokenize(chunk)
            emb = self.forward(ids, mask)
            out.extend(list(emb.cpu().numpy()))
        return out


_MODEL_CACHE: dict = {}


def get_encoder(device=None, cfg: EncoderConfig | None = None) -> NativeEncoder:
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    key = (str(device), tuple(vars(cfg or EncoderConfig()).items()))
    if key not in _MODEL_CACHE:
        _MODEL_CACHE[key] = NativeEncoder(cfg, device=device)
    return _MODEL_CACHE[key]
