"""Two-Tower retrieval end-to-end (BASELINE config 4 shape, CPU-sized):
query tower (causal transformer over the history) + item tower (embeddings
+ SwiGLU blocks), InfoNCE over in-batch negatives.  On a multi-GPU run the
negatives are shared across ranks with a differentiable RCCL all-gather
(replay_amd/parallel/collectives.py); on CPU/1 GPU the same code path
degrades to the local batch."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))  # repo root

import numpy as np
import pandas as pd
import torch

from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureType
from replay_amd.data.nn import (
    SequenceTokenizer,
    TensorFeatureInfo,
    TensorSchema,
    TorchSequentialDataset,
)
from replay_amd.nn.lightning import LightningModule, OptimizerFactory
from replay_amd.nn.sequential.twotower import TwoTower
from replay_amd.train import Trainer

N_ITEMS, MAX_LEN = 300, 20


def main():
    rng = np.random.default_rng(5)
    rows = []
    for user in range(400):
        base = rng.integers(0, N_ITEMS)
        for t in range(int(rng.integers(5, 25))):
            rows.append((user, (base + t * 7) % N_ITEMS, t))
    log = pd.DataFrame(rows, columns=["user_id", "item_id", "timestamp"])

    schema = FeatureSchema(
        [
            FeatureInfo("user_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
            FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
        ]
    )
    tensor_schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id", FeatureType.CATEGORICAL, is_seq=True,
                feature_hint=FeatureHint.ITEM_ID, cardinality=N_ITEMS, embedding_dim=32,
            )
        ]
    )
    tokenizer = SequenceTokenizer(tensor_schema)
    sequences = tokenizer.fit_transform(Dataset(feature_schema=schema, interactions=log))
    train_ds = TorchSequentialDataset(sequences, MAX_LEN)

    class NextItemLoader:
        """labels = the sequence shifted by one (next-item objective)."""

        def __init__(self, loader):
            self.loader = loader

        def __iter__(self):
            for batch in self.loader:
                b = dict(batch)
                b["labels"] = b["item_id"].roll(-1, dims=1)
                b["labels_padding_mask"] = b["padding_mask"]
                yield b

        def __len__(self):
            return len(self.loader)

    loader = NextItemLoader(torch.utils.data.DataLoader(train_ds, batch_size=64, shuffle=True))

    model = TwoTower.from_params(tensor_schema, max_sequence_length=MAX_LEN, embedding_dim=32)
    module = LightningModule(model, OptimizerFactory(lr=1e-3))
    trainer = Trainer(max_epochs=2, precision="bf16-mixed")
    trainer.fit(module, loader)
    print("train_loss:", trainer.logged_metrics["train_loss"])

    # retrieval: the item tower caches its catalog embeddings on first
    # inference call (invalidated automatically on weight updates)
    model.eval()
    batch = next(iter(torch.utils.data.DataLoader(train_ds, batch_size=4)))
    scores = model.forward_inference(dict(batch))
    print("top-5 retrieved items:\n", scores.topk(5, dim=-1).indices)


if __name__ == "__main__":
    main()
