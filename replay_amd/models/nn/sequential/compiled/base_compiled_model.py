"""Compiled inference models.

Parity with reference replay/models/nn/sequential/compiled/
(base_compiled_model.py:18-52 ``_compile_openvino``: ONNX -> ov.CompiledModel
with modes batch / one_query / dynamic_batch_size :12-16; SasRecCompiled
sasrec_compiled.py:20; Bert4RecCompiled).  OpenVINO does not exist in the
ROCm stack; the MI355X equivalent is a torch.jit-traced graph (static shapes,
fused eager ops, no Python dispatch) for CPU or GPU serving — same
compiled-vs-eager parity contract, tested the same way.
"""

from __future__ import annotations

from typing import Optional

import torch

MODE_BATCH = "batch"
MODE_ONE_QUERY = "one_query"
MODE_DYNAMIC = "dynamic_batch_size"


class _InferenceWrapper(torch.nn.Module):
    """Trace-friendly wrapper: tensors in -> logits out."""

    def __init__(self, model: torch.nn.Module, item_feature_name: str) -> None:
        super().__init__()
        self.model = model
        self.item_feature_name = item_feature_name

    def forward(self, item_id: torch.Tensor, padding_mask: torch.Tensor) -> torch.Tensor:
        return self.model.forward_inference(
            {self.item_feature_name: item_id, "padding_mask": padding_mask}
        )


class BaseCompiledModel:
    def __init__(
        self,
        model: torch.nn.Module,
        mode: str = MODE_BATCH,
        batch_size: Optional[int] = 32,
        max_seq_len: int = 50,
        device: str = "cpu",
        item_feature_name: str = "item_id",
    ) -> None:
        if mode not in (MODE_BATCH, MODE_ONE_QUERY, MODE_DYNAMIC):
            raise ValueError(f"Unknown mode {mode}")
        self.mode = mode
        self.batch_size = 1 if mode == MODE_ONE_QUERY else batch_size
        self.max_seq_len = max_seq_len
        self.device = device
        self.item_feature_name = item_feature_name
        model = model.to(device).eval()
        wrapper = _InferenceWrapper(model, item_feature_name).eval()
        example_b = self.batch_size or 2
        example = (
            torch.zeros(example_b, max_seq_len, dtype=torch.long, device=device),
            torch.ones(example_b, max_seq_len, dtype=torch.bool, device=device),
        )
        with torch.no_grad():
            self._compiled = torch.jit.trace(wrapper, example, check_trace=False)
            self._compiled = torch.jit.freeze(self._compiled)

    def predict(self, batch) -> torch.Tensor:
        items = batch[self.item_feature_name].to(self.device)
        mask = batch["padding_mask"].to(self.device)
        if self.mode != MODE_DYNAMIC and self.batch_size is not None:
            if items.shape[0] != self.batch_size:
                raise ValueError(
                    f"mode={self.mode} expects batch_size={self.batch_size}, got {items.shape[0]}"
                )
        with torch.no_grad():
            return self._compiled(items, mask)

    __call__ = predict

    @classmethod
    def compile(cls, model, **kwargs) -> "BaseCompiledModel":
        return cls(model, **kwargs)


class SasRecCompiled(BaseCompiledModel):
    """Compiled SASRec (reference sasrec_compiled.py:20)."""


class Bert4RecCompiled(BaseCompiledModel):
    """Compiled BERT4Rec."""
