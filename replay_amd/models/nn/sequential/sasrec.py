"""Ready-to-train SASRec with the legacy loss zoo.

Parity with reference replay/models/nn/sequential/sasrec/lightning.py:22:
loss types BCE full (:278) / BCE sampled (:310) / CE full (:335) / CE sampled
with log-correction (:357-381) / SCE (:383); ``_get_sampled_logits`` (:394-472)
with ``global_uniform`` vs ``inbatch`` strategies and ``negatives_sharing``
(one negative set shared batch-wide — here shared across the node's GPUs by
an RCCL all-gather); embedding resize APIs
(:493-568).  Composed over the new-generation body (the MI355X build keeps
one transformer implementation).
"""

from __future__ import annotations

from typing import Optional

import torch

from replay_amd.data.nn.schema import TensorSchema
from replay_amd.nn.lightning.module import LightningModule
from replay_amd.nn.lightning.optimizer import OptimizerFactory
from replay_amd.nn.loss import BCE, BCESampled, CE, CESampled, ScalableCrossEntropyLoss
from replay_amd.nn.sequential.sasrec.model import SasRec as _NewGenSasRec


def _make_loss(loss_type: str, loss_sample_count: Optional[int], vocab_size: int):
    loss_type = loss_type.upper()
    sampled = loss_sample_count is not None and loss_sample_count > 0
    if loss_type == "CE":
        return CESampled(log_correction=True, vocab_size=vocab_size) if sampled else CE()
    if loss_type == "BCE":
        return BCESampled() if sampled else BCE()
    if loss_type == "SCE":
        return ScalableCrossEntropyLoss()
    raise ValueError(f"Unknown loss type {loss_type}")


class SasRec(LightningModule):
    """Legacy-surface SASRec (reference lightning.py:22 constructor args)."""

    def __init__(
        self,
        tensor_schema: TensorSchema,
        max_seq_len: int = 200,
        hidden_size: int = 50,
        block_count: int = 2,
        head_count: int = 1,
        dropout_rate: float = 0.3,
        loss_type: str = "CE",
        loss_sample_count: Optional[int] = None,
        negative_sampling_strategy: str = "global_uniform",
        negatives_sharing: bool = False,
        learning_rate: float = 1e-3,
    ) -> None:
        if negative_sampling_strategy not in ("global_uniform", "inbatch"):
            raise ValueError("negative_sampling_strategy must be global_uniform/inbatch")
        self._schema = tensor_schema
        item_feature = tensor_schema.item_id_feature_name
        vocab = tensor_schema[item_feature].cardinality
        loss = _make_loss(loss_type, loss_sample_count, vocab)
        model = _NewGenSasRec.from_params(
            tensor_schema,
            max_sequence_length=max_seq_len,
            embedding_dim=hidden_size,
            num_blocks=block_count,
            num_heads=head_count,
            dropout=dropout_rate,
        )
        model.loss = loss
        model.loss.set_logits_callback(model.head)
        super().__init__(model, OptimizerFactory(lr=learning_rate))
        self.loss_type = loss_type
        self.loss_sample_count = loss_sample_count
        self.negative_sampling_strategy = negative_sampling_strategy
        self.negatives_sharing = negatives_sharing
        self._vocab_size = vocab
        self.hparams = {
            "max_seq_len": max_seq_len,
            "hidden_size": hidden_size,
            "block_count": block_count,
            "head_count": head_count,
            "dropout_rate": dropout_rate,
            "loss_type": loss_type,
            "loss_sample_count": loss_sample_count,
        }

    # -- sampled negatives (reference :394-472) --------------------------------
    def _sample_negatives(self, batch) -> Optional[torch.Tensor]:
        n = self.loss_sample_count
        if not n:
            return None
        device = batch["padding_mask"].device
        if self.negative_sampling_strategy == "inbatch":
            negatives = batch["labels"][batch.get("labels_padding_mask", batch["padding_mask"])]
            negatives = torch.unique(negatives.reshape(-1))
        else:
            negatives = torch.randint(0, self._vocab_size, (n,), device=device)
        if self.negatives_sharing:
            from replay_amd.parallel import gather_ids

            negatives = gather_ids(negatives)
        return negatives

    def training_step(self, batch, batch_idx: int = 0) -> torch.Tensor:
        if "negatives" not in batch:
            negatives = self._sample_negatives(batch)
            if negatives is not None:
                batch = dict(batch)
                batch["negatives"] = negatives
        loss = self._model(batch)
        self.log("train_loss", loss, sync_dist=True)
        return loss

    # -- embedding resize APIs (reference :493-568) ----------------------------
    @property
    def _item_embedder(self):
        return self._model.body.embedder.embedders[self._model.item_feature_name]

    def get_all_embeddings(self) -> dict:
        emb = self._item_embedder
        return {"item_embedding": emb.item_emb.weight.detach().clone()[: emb.cardinality]}

    def set_item_embeddings_by_size(self, new_size: int) -> None:
        """Grow the item-embedding table to ``new_size`` items, keeping
        learned rows (reference :493)."""
        emb = self._item_embedder
        old = emb.item_emb
        if new_size < emb.cardinality:
            raise ValueError("new_size must be >= current cardinality")
        n_special = old.num_embeddings - emb.cardinality  # pad + extras
        new_table = torch.nn.Embedding(new_size + n_special, old.embedding_dim, padding_idx=new_size)
        with torch.no_grad():
            new_table.weight[: emb.cardinality] = old.weight[: emb.cardinality]
        emb.item_emb = new_table
        emb.cardinality = new_size
        self._vocab_size = new_size
        self._schema[self._model.item_feature_name]._set_cardinality(new_size)

    def set_item_embeddings_by_tensor(self, tensor: torch.Tensor) -> None:
        """Replace item rows from a [n_items, E] tensor (reference :520)."""
        emb = self._item_embedder
        if tensor.shape[1] != emb.embedding_dim:
            raise ValueError("embedding_dim mismatch")
        if tensor.shape[0] != emb.cardinality:
            self.set_item_embeddings_by_size(tensor.shape[0])
            emb = self._item_embedder
        with torch.no_grad():
            emb.item_emb.weight[: tensor.shape[0]] = tensor

    def append_item_embeddings(self, tensor: torch.Tensor) -> None:
        """Append new item rows (reference :545)."""
        emb = self._item_embedder
        old_n = emb.cardinality
        self.set_item_embeddings_by_size(old_n + tensor.shape[0])
        with torch.no_grad():
            self._item_embedder.item_emb.weight[old_n : old_n + tensor.shape[0]] = tensor


# ---------------------------------------------------------------------------
# Legacy dataset/batch surface (reference models/nn/sequential/sasrec/
# dataset.py): thin adapters over the new-generation TorchSequentialDataset
# that emit the dict keys this legacy SasRec consumes.
# ---------------------------------------------------------------------------
from typing import Dict, NamedTuple  # noqa: E402

from replay_amd.data.nn import (  # noqa: E402
    TorchSequentialDataset,
    TorchSequentialValidationDataset,
)

SasRecModel = _NewGenSasRec  # the raw torch module (reference model.py SasRecModel)


class SasRecTrainingBatch(NamedTuple):
    """Legacy tuple view of a training batch (reference dataset.py:20)."""

    query_id: torch.LongTensor
    padding_mask: torch.BoolTensor
    features: Dict[str, torch.Tensor]
    labels: torch.LongTensor
    labels_padding_mask: torch.BoolTensor


class SasRecPredictionBatch(NamedTuple):
    """Legacy tuple view of a prediction batch (reference dataset.py:133)."""

    query_id: torch.LongTensor
    padding_mask: torch.BoolTensor
    features: Dict[str, torch.Tensor]

    def convert_to_dict(self) -> dict:
        out = {"query_id": self.query_id, "padding_mask": self.padding_mask}
        out.update(self.features)
        return out


class SasRecValidationBatch(NamedTuple):
    """Legacy tuple view of a validation batch (reference dataset.py:195)."""

    query_id: torch.LongTensor
    padding_mask: torch.BoolTensor
    features: Dict[str, torch.Tensor]
    ground_truth: torch.LongTensor
    train: torch.LongTensor


class SasRecTrainingDataset(torch.utils.data.Dataset):
    """Next-item training samples: the sequence shifted by ``sequence_shift``
    becomes the labels (reference dataset.py:43)."""

    def __init__(
        self,
        sequential,
        max_sequence_length: int,
        sequence_shift: int = 1,
        sliding_window_step=None,
        label_feature_name=None,
    ) -> None:
        self._shift = sequence_shift
        self._label_name = label_feature_name or sequential.schema.item_id_feature_name
        if label_feature_name is not None:
            feat = sequential.schema[label_feature_name]
            if not (feat.is_cat and feat.is_seq):
                raise ValueError("Label feature must be a categorical sequence")
        self._seq_names = [n for n, f in sequential.schema.items() if f.is_seq]
        self._inner = TorchSequentialDataset(
            sequential,
            max_sequence_length + sequence_shift,
            sliding_window_step=sliding_window_step,
        )

    def __len__(self) -> int:
        return len(self._inner)

    def __getitem__(self, index: int) -> dict:
        item = dict(self._inner[index])
        labels = item[self._label_name][self._shift :]
        labels_padding_mask = item["padding_mask"][self._shift :]
        for name in self._seq_names:
            item[name] = item[name][: -self._shift]
        item["padding_mask"] = item["padding_mask"][: -self._shift]
        item["labels"] = labels
        item["labels_padding_mask"] = labels_padding_mask
        return item


class SasRecPredictionDataset(torch.utils.data.Dataset):
    """Inference samples: the full (left-padded) history (reference
    dataset.py:150)."""

    def __init__(self, sequential, max_sequence_length: int, padding_value=None) -> None:
        self._inner = TorchSequentialDataset(sequential, max_sequence_length)

    def __len__(self) -> int:
        return len(self._inner)

    def __getitem__(self, index: int) -> dict:
        return dict(self._inner[index])


class SasRecValidationDataset(torch.utils.data.Dataset):
    """Validation samples carrying ground_truth + train ids (reference
    dataset.py:212)."""

    def __init__(self, sequential, ground_truth, train, max_sequence_length: int, label_feature_name=None) -> None:
        self._inner = TorchSequentialValidationDataset(
            sequential, ground_truth=ground_truth, train=train,
            max_sequence_length=max_sequence_length, label_feature_name=label_feature_name,
        )

    def __len__(self) -> int:
        return len(self._inner)

    def __getitem__(self, index: int) -> dict:
        return dict(self._inner[index])
