"""BERT4Rec training end-to-end (the reference's examples/10 flow):
masked-token training on bidirectional attention, then append-mask
inference for next-item prediction.  Runs on CPU; on an MI355X the
bidirectional MFMA attention kernel and the fused linear+CE pair engage
automatically."""

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
from replay_amd.nn.sequential.bert4rec import Bert4Rec
from replay_amd.nn.transform import TokenMaskTransform
from replay_amd.train import Trainer

N_ITEMS, MAX_LEN = 400, 24


def main():
    rng = np.random.default_rng(3)
    rows = []
    for user in range(400):
        start = rng.integers(0, N_ITEMS)
        for t in range(int(rng.integers(6, 30))):
            rows.append((user, (start + 3 * t) % N_ITEMS, t))
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
                feature_hint=FeatureHint.ITEM_ID, cardinality=N_ITEMS, embedding_dim=64,
            )
        ]
    )
    tokenizer = SequenceTokenizer(tensor_schema)
    sequences = tokenizer.fit_transform(Dataset(feature_schema=schema, interactions=log))
    train_ds = TorchSequentialDataset(sequences, MAX_LEN)

    # BERT4Rec trains on randomly masked tokens (reference bert4rec/dataset.py)
    mask_transform = TokenMaskTransform(mask_prob=0.2, generator_seed=0)

    class MaskedLoader:
        def __init__(self, loader):
            self.loader = loader

        def __iter__(self):
            for batch in self.loader:
                b = dict(batch)
                b["labels"] = b["item_id"].clone()
                b["labels_padding_mask"] = b["padding_mask"]
                yield mask_transform(b)

        def __len__(self):
            return len(self.loader)

    loader = MaskedLoader(torch.utils.data.DataLoader(train_ds, batch_size=64, shuffle=True))

    model = Bert4Rec.from_params(
        tensor_schema, max_sequence_length=MAX_LEN, embedding_dim=64, num_blocks=2, num_heads=2
    )
    module = LightningModule(model, OptimizerFactory(lr=1e-3))
    trainer = Trainer(max_epochs=2, precision="bf16-mixed")
    trainer.fit(module, loader)
    print("train_loss:", trainer.logged_metrics["train_loss"])

    # inference: the model appends the mask token after the last real item
    # (reference convention) and scores the catalog at that position
    model.eval()
    batch = next(iter(torch.utils.data.DataLoader(train_ds, batch_size=4)))
    logits = model.forward_inference(dict(batch))
    top10 = logits.topk(10, dim=-1).indices
    print("next-item top-10 for 4 users:\n", top10)


if __name__ == "__main__":
    main()
