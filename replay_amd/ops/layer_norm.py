"""LayerNorm dispatch: fused HIP kernel on GPU, torch eager on CPU.

The HIP kernel (K3 in SURVEY §2.12) is a wave-per-row fused
LayerNorm(+residual) for bf16/fp32 rows up to a few thousand elements.
"""

from __future__ import annotations

import torch

from replay_amd.ops import hip_ext, require_hip_on_gpu


class LayerNorm(torch.nn.LayerNorm):
    """Drop-in LayerNorm that routes to the HIP fused kernel on GPU."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and require_hip_on_gpu(x):
            ext = hip_ext()
            if (
                hasattr(ext, "layer_norm_fwd")
                and x.dtype in (torch.bfloat16, torch.float16, torch.float32)
                and x.shape[-1] == self.normalized_shape[-1]
            ):
                from replay_amd.ops.autograd import LayerNormFunction

                return LayerNormFunction.apply(x, self.weight, self.bias, self.eps)
        return super().forward(x)
