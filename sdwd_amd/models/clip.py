"""CLIP text encoder (the conditioning stage the reference delegated to the
remote webui's CLIP via POST /txt2img — SURVEY.md §2.5).

A ViT-L/14-text-shaped transformer: vocab 49408, 77 positions, causal mask,
quick-GELU MLPs, final LayerNorm. SDXL adds a second, bigger encoder
(penultimate-layer output + pooled embedding) — see registry._build_sdxl.

Weights are random-init (no network in this environment); the architecture
and shapes match SD1.5 so the benchmark works the same compute.
"""
from __future__ import annotations

import torch
import torch.nn as nn




class CLIPAttention(nn.Module):
    def __init__(self, d_model: int, heads: int):
        super().__init__()
        self.heads = heads
        self.d_head = d_model // heads
        self.qkv = nn.Linear(d_model, 3 * d_model)
        self.out = nn.Linear(d_model, d_model)

    def forward(self, x: torch.Tensor, causal_bias: torch.Tensor) -> torch.Tensor:
        b, s, d = x.shape
        qkv = self.qkv(x).view(b, s, 3, self.heads, self.d_head)
        q, k, v = qkv.unbind(dim=2)  # [b, s, h, dh]
        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        # 77-token causal attention: tiny; masked matmul-softmax path.
        scale = self.d_head**-0.5
        attn = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
        attn = attn + causal_bias
        attn = attn.softmax(dim=-1).to(v.dtype)
        out = torch.matmul(attn, v)
        out = out.transpose(1, 2).reshape(b, s, d)
        return self.out(out)


class QuickGELU(nn.Module):
    def forward(self, x):
        return x * torch.sigmoid(1.702 * x)


class CLIPBlock(nn.Module):
    def __init__(self, d_model: int, heads: int):
        super().__init__()
        self.ln1 = nn.LayerNorm(d_model)
        self.attn = CLIPAttention(d_model, heads)
        self.ln2 = nn.LayerNorm(d_model)
        self.mlp = nn.Sequential(
            nn.Linear(d_model, 4 * d_model),
            QuickGELU(),
            nn.Linear(4 * d_model, d_model),
        )

    def forward(self, x, causal_bias):
        x = x + self.attn(self.ln1(x), causal_bias)
        x = x + self.mlp(self.ln2(x))
        return x


class CLIPTextEncoder(nn.Module):
    def __init__(
        self,
        vocab_size: int = 49408,
        max_len: int = 77,
        d_model: int = 768,
        layers: int = 12,
        heads: int = 12,
    ):
        super().__init__()
        self.max_len = max_len
        self.d_model = d_model
        self.token_emb = nn.Embedding(vocab_size, d_model)
        self.pos_emb = nn.Parameter(torch.zeros(max_len, d_model))
        self.blocks = nn.ModuleList(CLIPBlock(d_model, heads) for _ in range(layers))
        self.ln_final = nn.LayerNorm(d_model)
        mask = torch.full((max_len, max_len), float("-inf")).triu(1)
        self.register_buffer("causal_bias", mask, persistent=False)
        # open_clip text encoders project the pooled EOT embedding
        # (pooled = h[eot] @ text_projection); absent for SD1.5-style CLIP
        self.text_proj: torch.nn.Parameter | None = None

    def set_text_projection(self, weight: torch.Tensor) -> None:
        self.text_proj = nn.Parameter(weight.clone())  # auto-registers

    def forward(
        self,
        tokens: torch.Tensor,
        penultimate: bool = False,
        clip_skip: int = 1,
    ) -> torch.Tensor:
        """tokens: [B, 77] int64 -> [B, 77, d_model] conditioning.

        ``penultimate``: SDXL conditioning — skip the last block and do NOT
        apply the final LayerNorm (sgm "penultimate" convention).
        ``clip_skip``: sdwui CLIP_stop_at_last_layers — skip the last
        ``clip_skip - 1`` blocks but DO apply the final LayerNorm afterwards
        (sd_hijack_clip semantics)."""
        safe = tokens.clamp_max(self.token_emb.num_embeddings - 1)
        x = self.token_emb(safe) + self.pos_emb
        # textual-inversion placeholders (ids >= vocab) -> trained vectors
        from .embeddings import apply_to_hidden

        x = apply_to_hidden(tokens, x)
        bias = self.causal_bias.to(x.dtype)
        skip = 1 if penultimate else max(0, clip_skip - 1)
        n = len(self.blocks) - skip
        for blk in self.blocks[:n]:
            x = blk(x, bias)
        if not penultimate:
            x = self.ln_final(x)
        return x

    def pooled(self, tokens: torch.Tensor, hidden: torch.Tensor) -> torch.Tensor:
        """EOT-token pooled embedding (SDXL conditioning)."""
        from .embeddings import PLACEHOLDER_BASE

        tok = tokens.clone()
        tok[tok >= PLACEHOLDER_BASE] = 0  # TI placeholders are not EOT
        eot = tok.argmax(dim=-1)  # highest id = end-of-text
        h = hidden[torch.arange(hidden.shape[0]), eot]
        if self.text_proj is not None:
            h = h @ self.text_proj.to(h.dtype)
        return h
