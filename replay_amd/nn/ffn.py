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
        self.w1 = torch.nn.Linear(embedding_dim, hidden)
        self.w2 = torch.nn.Linear(hidden, embedding_dim)
        self.dropout1 = torch.nn.Dropout(dropout)
        self.dropout2 = torch.nn.Dropout(dropout)
        self.activation = torch.nn.ReLU() if activation == "relu" else torch.nn.GELU()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.dropout2(self.w2(self.dropout1(self.activation(self.w1(x)))))
        return out + x


class SwiGLU(torch.nn.Module):
    """Gated MLP: W2(silu(W1 x) * W3 x) (reference ffn.py:60)."""

    def __init__(self, embedding_dim: int, hidden_dim: int = None, dropout: float = 0.0) -> None:
        super().__init__()
        hidden = hidden_dim or embedding_dim * 4
        self.w1 = torch.nn.Linear(embedding_dim, hidden, bias=False)
        self.w3 = torch.nn.Linear(embedding_dim, hidden, bias=False)
        self.w2 = torch.nn.Linear(hidden, embedding_dim, bias=False)
        self.dropout = torch.nn.Dropout(dropout)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.dropout(self.w2(torch.nn.functional.silu(self.w1(x)) * self.w3(x)))


class SwiGLUEncoder(torch.nn.Module):
    """SwiGLU + RMSNorm block stack for the Two-Tower item tower
    (reference ffn.py:102)."""

    def __init__(self, embedding_dim: int, hidden_dim: int = None, num_blocks: int = 1, dropout: float = 0.0) -> None:
        super().__init__()
        self.blocks = torch.nn.ModuleList(
            [SwiGLU(embedding_dim, hidden_dim, dropout) for _ in range(num_blocks)]
        )
        self.norms = torch.nn.ModuleList(
            [torch.nn.RMSNorm(embedding_dim) for _ in range(num_blocks)]
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for block, norm in zip(self.blocks, self.norms):
            x = x + block(norm(x))
        return x
