"""Llama-3-style decoder-only transformer (random-init, synthetic-data
training) for the DP bucket-fusion benchmark.

Standard architecture: RMSNorm, RoPE, GQA attention (SDPA), SwiGLU MLP,
untied small-vocab-friendly embedding. Written for bf16 training on MI355X
(288 GB HBM3E fits the 8B config with Adam on one GPU).
"""
import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    dim: int = 4096
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    ffn_dim: int = 14336
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5


def llama3_8b() -> LlamaConfig:
    return LlamaConfig()


def llama_debug() -> LlamaConfig:
    return LlamaConfig(vocab_size=1024, dim=256, n_layers=4, n_heads=8,
                       n_kv_heads=4, ffn_dim=688, max_seq_len=512)


class RMSNorm(nn.Module):
    def __init__(self, dim, eps):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x):
        if x.is_cuda:
            from ..ops.rmsnorm import fused_rmsnorm, fused_rmsnorm_supported
            xc = x.contiguous()
            if self.weight.dtype == torch.float32 and fused_rmsnorm_supported(xc):
                return fused_rmsnorm(xc, self.weight, self.eps)
        dt = x.dtype
        x = x.float()
        x = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + self.eps)
        return (x * self.weight.float()).to(dt)


def _add_norm(norm: RMSNorm, res, branch):
    """(normed, new_res) = (rmsnorm(res + branch), res + branch); the add is
    fused into the norm kernel on bf16 CUDA (csrc/kernels/rmsnorm.hip) —
    the residual stream never takes a separate elementwise pass. branch may
    be None (start of the stream)."""
    if branch is None:
        return norm(res), res
    if res.is_cuda:
        from ..ops.rmsnorm import fused_add_rmsnorm, fused_rmsnorm_supported
        rc = res.contiguous()
        bc = branch.contiguous()
        if (norm.weight.dtype == torch.float32 and bc.dtype == rc.dtype
                and fused_rmsnorm_supported(rc)):
            return fused_add_rmsnorm(rc, bc, norm.weight, norm.eps)
    h = res + branch
    return norm(h), h


def _rope_freqs(dim, max_seq, theta, device):
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, device=device).float() / dim))
    t = torch.arange(max_seq, device=device).float()
    freqs = torch.outer(t, inv)
    return torch.polar(torch.ones_like(freqs), freqs)  # complex64 [T, dim/2]


def _apply_rope(x, freqs):
    # x: [B, T, H, D]
    b, t, h, d = x.shape
    xc = torch.view_as_complex(x.float().reshape(b, t, h, d // 2, 2))
    out = torch.view_as_real(xc * freqs[:t].view(1, t, 1, d // 2))
    return out.reshape(b, t, h, d).type_as(x)


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.n_heads = cfg.n_heads
        self.n_kv = cfg.n_kv_heads
        self.hd = cfg.dim // cfg.n_heads
        self.wq = nn.Linear(cfg.dim, cfg.n_heads * self.hd, bias=False)
        self.wk = nn.Linear(cfg.dim, cfg.n_kv_heads * self.hd, bias=False)
        self.wv = nn.Linear(cfg.dim, cfg.n_kv_heads * self.hd, bias=False)
        self.wo = nn.Linear(cfg.n_heads * self.hd, cfg.dim, bias=False)

    def forward(self, x, freqs):
        b, t, _ = x.shape
        q = self.wq(x).view(b, t, self.n_heads, self.hd)
        k = self.wk(x).view(b, t, self.n_kv, self.hd)
        v = self.wv(x).view(b, t, self.n_kv, self.hd)
        q = _apply_rope(q, freqs)
        k = _apply_rope(k, freqs)
        q, k, v = (z.transpose(1, 2) for z in (q, k, v))  # [B, H, T, D]
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                           enable_gqa=self.n_kv != self.n_heads)
        return self.wo(o.transpose(1, 2).reshape(b, t, -1))


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.w1 = nn.Linear(cfg.dim, cfg.ffn_dim, bias=False)
        self.w3 = nn.Linear(cfg.dim, cfg.ffn_dim, bias=False)
        self.w2 = nn.Linear(cfg.ffn_dim, cfg.dim, bias=False)

    def forward(self, x):
        return self.w2(F.silu(self.w1(x)) * self.w3(x))


class Block(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn = Attention(cfg)
        self.mlp = MLP(cfg)
        self.ln1 = RMSNorm(cfg.dim, cfg.norm_eps)
        self.ln2 = RMSNorm(cfg.dim, cfg.norm_eps)

    def forward(self, res, branch, freqs):
        """Carries the residual stream as a (res, pending-branch) pair so the
        pre-norm adds fuse into the norm kernels: equivalent to
        x = x + attn(ln1(x)); x = x + mlp(ln2(x))."""
        n1, res = _add_norm(self.ln1, res, branch)
        a = self.attn(n1, freqs)
        n2, res = _add_norm(self.ln2, res, a)
        return res, self.mlp(n2)


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.head = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
        self.register_buffer("freqs", torch.empty(0), persistent=False)
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=0.02 / math.sqrt(2 * cfg.n_layers))
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=0.02)

    def forward(self, tokens, targets=None):
        if self.freqs.numel() == 0 or self.freqs.device != tokens.device:
            self.freqs = _rope_freqs(self.cfg.dim // self.cfg.n_heads,
                                     self.cfg.max_seq_len, self.cfg.rope_theta,
                                     tokens.device)
        x = self.embed(tokens)
        if x.is_cuda and torch.is_autocast_enabled():
            # bf16 residual stream (standard bf16 mixed precision; RMSNorm
            # stats still accumulate in fp32 inside the fused kernel).
            x = x.to(torch.bfloat16)
        branch = None
        for blk in self.blocks:
            x, branch = blk(x, branch, self.freqs)
        x, _ = _add_norm(self.norm, x, branch)
        if targets is not None:
            logits = self.head(x)
            return F.cross_entropy(logits.float().view(-1, logits.size(-1)),
                                   targets.view(-1))
        return self.head(x)
