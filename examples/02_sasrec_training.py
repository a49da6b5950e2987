"""SASRec training end-to-end (the reference's examples/09 flow):
interactions -> SequenceTokenizer -> torch datasets -> Trainer ->
validation metrics -> top-k predictions with filter_seen.
Runs on CPU or GPU (bf16 + HIP kernels when on MI355X)."""

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
    TorchSequentialValidationDataset,
)
from replay_amd.nn.lightning import (
    ComputeMetricsCallback,
    LightningModule,
    OptimizerFactory,
    PandasTopItemsCallback,
    SeenItemsFilter,
)
from replay_amd.nn.sequential.sasrec import SasRec
from replay_amd.nn.transform import make_default_sasrec_transforms
from replay_amd.train import Trainer

N_ITEMS, MAX_LEN = 500, 30


def main():
    rng = np.random.default_rng(0)
    rows = []
    for user in range(500):
        start = rng.integers(0, N_ITEMS)
        for t in range(int(rng.integers(5, 40))):
            rows.append((user, (start + t) % N_ITEMS, t))
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
    transforms = make_default_sasrec_transforms(tensor_schema)

    class TransformLoader:
        def __init__(self, loader, pipeline):
            self.loader, self.pipeline = loader, pipeline

        def __iter__(self):
            for batch in self.loader:
                yield self.pipeline(dict(batch))

        def __len__(self):
            return len(self.loader)

    loader = TransformLoader(torch.utils.data.DataLoader(train_ds, batch_size=64, shuffle=True), transforms["train"])

    model = SasRec.from_params(tensor_schema, max_sequence_length=MAX_LEN, embedding_dim=64, num_blocks=2, num_heads=2)
    module = LightningModule(model, OptimizerFactory(lr=1e-3))
    trainer = Trainer(max_epochs=3, precision="bf16-mixed")
    trainer.fit(module, loader)
    print("train_loss:", trainer.logged_metrics["train_loss"])

    # validation metrics
    val_ds = TorchSequentialValidationDataset(sequences, sequences, max_sequence_length=MAX_LEN)
    metrics_cb = ComputeMetricsCallback(metrics=["ndcg", "recall"], top_k=[1, 10], item_count=N_ITEMS)
    Trainer(callbacks=[metrics_cb], precision="bf16-mixed").validate(
        module, torch.utils.data.DataLoader(val_ds, batch_size=64)
    )
    print("metrics:", metrics_cb.metric_history[-1])

    # top-k recommendations with filter_seen
    recs_cb = PandasTopItemsCallback(top_k=10, postprocessors=[SeenItemsFilter()], query_column="query_id")
    Trainer(callbacks=[recs_cb], precision="bf16-mixed").predict(
        module, torch.utils.data.DataLoader(train_ds, batch_size=64), return_predictions=False
    )
    print(recs_cb.get_result().head())


if __name__ == "__main__":
    main()
