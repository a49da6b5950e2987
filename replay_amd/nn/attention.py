"""Attention modules.

``MultiheadAttention`` here is this framework's own implementation (same math
as the reference's use of torch.nn.MultiheadAttention inside
replay/nn/sequential/sasrec/transformer.py:38,99-106): QKV projection GEMMs +
scores + additive mask + softmax + PV + out projection.  It is written so the
scores->softmax->PV middle runs through replay_amd.ops (HIP flash-style
fused attention on GPU, K1/K2 in SURVEY §2.12).

``MultiHeadDifferentialAttention`` gives parity with reference
replay/nn/attention.py:7 (dual softmax attentions subtracted with a learned
lambda, per-head RMSNorm, reference :67-157).
"""

from __future__ import annotations

import math
from typing import Optional

import torch


def _attention_core(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    attn_mask: Optional[torch.Tensor],
    dropout_p: float,
    training: bool,
) -> torch.Tensor:
    """scores -> +mask -> softmax -> PV.  q,k,v: [B*H, L, Dh];
    attn_mask additive float [B*H, L, L] or None."""
    from replay_amd.ops.attention import fused_attention

    return fused_attention(q, k, v, attn_mask, dropout_p if training else 0.0)


class MultiheadAttention(torch.nn.Module):
    """Batch-first multi-head self-attention with additive float mask."""

    def __init__(self, embed_dim: int, num_heads: int, dropout: float = 0.0, bias: bool = True) -> None:
        super().__init__()
        if embed_dim % num_heads != 0:
            raise ValueError("embed_dim must divide num_heads")
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.dropout = dropout
        from replay_amd.ops.fast_linear import ChunkedWgradLinear

        self.in_proj = ChunkedWgradLinear(embed_dim, 3 * embed_dim, bias=bias)
        self.out_proj = ChunkedWgradLinear(embed_dim, embed_dim, bias=bias)

    def _can_use_flash(self, x: torch.Tensor, attn_mask) -> bool:
        from replay_amd.nn.mask import MaskSpec
        from replay_amd.ops import hip_ext

        if not isinstance(attn_mask, MaskSpec) or not x.is_cuda:
            return False
        if self.dropout > 0.0 and self.training:
            return False
        if x.dtype not in (torch.bfloat16, torch.float16, torch.float32):
            return False
        ext = hip_ext()
        if ext is None or not hasattr(ext, "attention_fwd"):
            return False
        L, Dh = x.shape[1], self.head_dim
        if Dh > 64 or 256 % Dh != 0 or L > 512:
            return False
        # the MFMA backward (bf16, Dh 32/64, L<=256) has a much smaller LDS
        # footprint than the VALU fallback — gate on the kernel that will
        # actually run (the VALU formula wrongly rejected bert4rec's
        # L=200/Dh=32 shape: 168 KB vs the MFMA path's 52 KB)
        runs_bf16 = x.dtype == torch.bfloat16 or torch.is_autocast_enabled()
        if runs_bf16 and Dh in (32, 64) and L <= 256:
            lpad = (L + 31) & ~31
            lds_bwd = 3 * Dh * lpad * 2 + lpad * 4 + 8 * 512 * 2 + L + 64
            return lds_bwd <= 160 * 1024
        # backward LDS budget check (fp32 staging): D*Lpad + 5*L*D + extras
        lpad = (L + 63) & ~63
        lds_bwd = 4 * (Dh * lpad + 5 * L * Dh + L + 8 * L) + L
        return lds_bwd <= 160 * 1024

    def forward(
        self,
        x: torch.Tensor,
        attn_mask=None,
        key_padding_mask: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """x: [B, L, E]; attn_mask: [B*H, L, L] additive float or a MaskSpec;
        key_padding_mask: [B, L] bool, True = PAD (torch convention)."""
        from replay_amd.nn.mask import MaskSpec

        B, L, E = x.shape
        H, Dh = self.num_heads, self.head_dim
        qkv = self.in_proj(x)  # [B, L, 3E]
        q, k, v = qkv.chunk(3, dim=-1)

        if key_padding_mask is None and self._can_use_flash(x, attn_mask):
            from replay_amd.ops.autograd import FlashAttentionFunction

            def split4(t):
                return t.view(B, L, H, Dh).transpose(1, 2).contiguous()

            out = FlashAttentionFunction.apply(
                split4(q), split4(k), split4(v), attn_mask.padding_mask, attn_mask.causal
            )
            out = out.transpose(1, 2).reshape(B, L, E)
            return self.out_proj(out)

        if isinstance(attn_mask, MaskSpec):
            attn_mask = attn_mask.materialize()

        def split(t):
            return t.view(B, L, H, Dh).transpose(1, 2).reshape(B * H, L, Dh)

        q, k, v = split(q), split(k), split(v)
        if key_padding_mask is not None:
            kp = torch.zeros(B, L, dtype=q.dtype, device=q.device)
            fill = float("-inf") if self.training else torch.finfo(torch.float32).min
            kp = kp.masked_fill(key_padding_mask, fill)
            kp = kp[:, None, None, :].expand(B, H, L, L).reshape(B * H, L, L)
            attn_mask = kp if attn_mask is None else attn_mask + kp
        out = _attention_core(q, k, v, attn_mask, self.dropout, self.training)
        out = out.view(B, H, L, Dh).transpose(1, 2).reshape(B, L, E)
        return self.out_proj(out)


class RMSNormPerHead(torch.nn.Module):
    def __init__(self, head_dim: int, eps: float = 1e-5) -> None:
        super().__init__()
        self.eps = eps
        self.weight = torch.nn.Parameter(torch.ones(head_dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        norm = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + self.eps)
        return norm * self.weight


class MultiHeadDifferentialAttention(torch.nn.Module):
    """Differential attention (reference replay/nn/attention.py:7-157).

    Parameter layout matches the reference exactly for state-dict
    compatibility: ``W_q``/``W_k`` project E -> 2*qk_e_head*num_heads (the
    per-head q/k split into two components of full head width), ``W_v``
    projects E -> v_e_head*num_heads, lambda parameters are per-head of
    shape ``(num_heads, qk_e_head)``, and the per-head RMSNorm scale is
    ``rms_scale`` of shape ``(v_e_head,)``.
    """

    def __init__(
        self,
        embedding_dim: int,
        num_heads: int,
        lambda_init: float,
        bias: bool = False,
        kdim: Optional[int] = None,
        vdim: Optional[int] = None,
    ) -> None:
        super().__init__()
        kdim = kdim or embedding_dim
        vdim = vdim or embedding_dim
        if kdim % num_heads != 0 or vdim % num_heads != 0:
            raise ValueError("kdim and vdim must be divisible by num_heads")
        self.qk_e_head = kdim // num_heads
        self.v_e_head = vdim // num_heads
        self.num_heads = num_heads
        self.lambda_init = lambda_init
        self.eps = 1e-5

        self.W_q = torch.nn.Linear(embedding_dim, 2 * self.qk_e_head * num_heads, bias=bias)
        self.W_k = torch.nn.Linear(embedding_dim, 2 * self.qk_e_head * num_heads, bias=bias)
        self.W_v = torch.nn.Linear(embedding_dim, self.v_e_head * num_heads, bias=bias)
        self.W_o = torch.nn.Linear(self.v_e_head * num_heads, embedding_dim, bias=bias)

        self.lambda_q1 = torch.nn.Parameter(torch.randn(num_heads, self.qk_e_head))
        self.lambda_k1 = torch.nn.Parameter(torch.randn(num_heads, self.qk_e_head))
        self.lambda_q2 = torch.nn.Parameter(torch.randn(num_heads, self.qk_e_head))
        self.lambda_k2 = torch.nn.Parameter(torch.randn(num_heads, self.qk_e_head))
        self.register_buffer(
            "scaling", torch.asarray(1.0 / math.sqrt(self.qk_e_head), dtype=torch.float32)
        )
        self.rms_scale = torch.nn.Parameter(torch.ones(self.v_e_head))

    def reset_parameters(self) -> None:
        for _, param in self.named_parameters():
            if param.dim() >= 2:
                torch.nn.init.xavier_normal_(param.data)

    def forward(
        self,
        query: torch.Tensor,
        key: torch.Tensor,
        value: torch.Tensor,
        attn_mask: torch.Tensor,
    ) -> torch.Tensor:
        """query/key/value: [B, L, E]; attn_mask additive float, shape
        [B*H, L, L] or [B, H, L, L] (0 = keep, -inf = masked)."""
        from replay_amd.nn.mask import MaskSpec

        if isinstance(attn_mask, MaskSpec):
            attn_mask = attn_mask.materialize()
        B, L, _ = value.shape
        H, Dq, Dv = self.num_heads, self.qk_e_head, self.v_e_head
        if attn_mask is None:
            attn_mask = torch.zeros(B, H, L, L, dtype=torch.float32, device=value.device)
        q = self.W_q(query).view(B, L, H, 2 * Dq).transpose(1, 2)  # [B, H, L, 2Dq]
        k = self.W_k(key).view(B, L, H, 2 * Dq).transpose(1, 2)
        v = self.W_v(value).view(B, L, H, Dv).transpose(1, 2)  # [B, H, L, Dv]
        q1, q2 = q.chunk(2, dim=-1)
        k1, k2 = k.chunk(2, dim=-1)

        # per-head lambda reparameterization: (H,)
        lam = (
            torch.exp((self.lambda_q1 * self.lambda_k1).sum(-1).float())
            - torch.exp((self.lambda_q2 * self.lambda_k2).sum(-1).float())
            + self.lambda_init
        ).view(1, H, 1, 1)

        if attn_mask.dim() == 3:
            attn_mask = attn_mask.view(B, H, L, L)
        scale = self.get_buffer("scaling")
        s1 = q1 @ k1.transpose(-2, -1) * scale + attn_mask
        s2 = q2 @ k2.transpose(-2, -1) * scale + attn_mask
        attn = torch.softmax(s1, dim=-1) - lam.to(s1.dtype) * torch.softmax(s2, dim=-1)
        out = attn @ v  # [B, H, L, Dv]

        # per-head RMSNorm + (1 - lambda_init) scaling
        rms = torch.sqrt(out.pow(2).mean(dim=-1, keepdim=True) + self.eps)
        out = (out / rms) * self.rms_scale * (1.0 - self.lambda_init)
        out = out.transpose(1, 2).contiguous().view(B, L, H * Dv)
        return self.W_o(out)
