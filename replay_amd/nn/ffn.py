"""Feed-forward blocks.

Parity with reference replay/nn/ffn.py (PointWiseFeedForward:11 —
conv1d(k=1) -> activation -> conv1d(k=1) + residual + dropout; SwiGLU:60;
SwiGLUEncoder:102).

MI355X note: a conv1d with kernel 1 IS a GEMM; both matmuls here map to MFMA
through hipBLASLt, and the activation+residual epilogue is fused by the HIP
fused-FFN kernel (K4/K5 in SURVEY §2.12) on the GPU path.
"""

from __future__ import annotations

import torch


class PointWiseFeedForward(torch.nn.Module):
    def __init__(
        self,
        embedding_dim: int,
        dropout: float = 0.0,
        activation: str = "relu",
        hidden_dim: int = None,
    ) -> None:
        super().__init__()
        hidden = hidden_dim or embedding_dim
        if activation not in ("relu", "gelu"):
            raise ValueError("activation must be relu or gelu")
        from replay_amd.ops.fast_linear import ChunkedWgradLinear

        self.w1 = ChunkedWgradLinear(embedding_dim, hidden)
        self.w2 = ChunkedWgradLinear(hidden, embedding_dim)
        self.dropout1 = torch.nn.Dropout(dropout)
        self.dropout2 = torch.nn.Dropout(dropout)
        self.activation = torch.nn.ReLU() if activation == "relu" else torch.nn.GELU()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.dropout2(self.w2(self.dropout1(self.activation(self.w1(x)))))
        return out + x


class SwiGLU(torch.nn.Module):
    """Gated MLP: W2(silu(WG x) * W1 x) — parameter names/shapes match
    reference ffn.py:60 for state-dict compatibility."""

    def __init__(self, embedding_dim: int, hidden_dim: int = None, dropout: float = 0.0) -> None:
        super().__init__()
        hidden = hidden_dim or embedding_dim * 2
        from replay_amd.ops.fast_linear import ChunkedWgradLinear

        self.WG = ChunkedWgradLinear(embedding_dim, hidden)
        self.W1 = ChunkedWgradLinear(embedding_dim, hidden)
        self.W2 = ChunkedWgradLinear(hidden, embedding_dim)
        self.dropout = torch.nn.Dropout(dropout)

    def reset_parameters(self) -> None:
        for _, param in self.named_parameters():
            if param.dim() >= 2:
                torch.nn.init.xavier_normal_(param.data)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.dropout(self.W2(torch.nn.functional.silu(self.WG(x)) * self.W1(x)))


class SwiGLUEncoder(torch.nn.Module):
    """Two post-norm SwiGLU blocks with skip connections for the Two-Tower
    item tower (reference ffn.py:102: x = norm(sw(x) + x), twice)."""

    def __init__(self, embedding_dim: int, hidden_dim: int = None, num_blocks: int = 2, dropout: float = 0.0) -> None:
        super().__init__()
        self.num_blocks = num_blocks
        for i in range(1, num_blocks + 1):
            setattr(self, f"sw{i}", SwiGLU(embedding_dim, hidden_dim, dropout))
            setattr(self, f"norm{i}", torch.nn.RMSNorm(embedding_dim))

    def reset_parameters(self) -> None:
        for i in range(1, self.num_blocks + 1):
            getattr(self, f"sw{i}").reset_parameters()
            getattr(self, f"norm{i}").reset_parameters()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for i in range(1, self.num_blocks + 1):
            x = getattr(self, f"norm{i}")(getattr(self, f"sw{i}")(x) + x)
        return x
