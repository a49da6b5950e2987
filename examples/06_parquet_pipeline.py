"""Production data path: tokenized sequences stored as parquet, streamed
through ParquetDataset/ParquetModule into the Trainer (the reference's
ParquetModule flow).  Under torchrun the same code shards rows per rank
with the rank-strided Partitioning (shared-seed shuffle)."""

import sys
import tempfile
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))  # repo root

import numpy as np
import pandas as pd
import torch

from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
from replay_amd.data.nn.parquet import ParquetModule
from replay_amd.data.schema import FeatureHint, FeatureType
from replay_amd.nn.lightning import LightningModule, OptimizerFactory
from replay_amd.nn.sequential.sasrec import SasRec
from replay_amd.nn.transform import make_default_sasrec_transforms
from replay_amd.train import Trainer

N_ITEMS, MAX_LEN = 200, 12


def write_sequences(path: Path, n_users: int = 300) -> None:
    rng = np.random.default_rng(0)
    rows = []
    for q in range(n_users):
        length = int(rng.integers(3, MAX_LEN + 1))
        rows.append({"query_id": q, "item_id": rng.integers(0, N_ITEMS, length).tolist()})
    pd.DataFrame(rows).to_parquet(path, index=False)


def main():
    tensor_schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id", FeatureType.CATEGORICAL, is_seq=True,
                feature_hint=FeatureHint.ITEM_ID, cardinality=N_ITEMS, embedding_dim=32,
            )
        ]
    )
    with tempfile.TemporaryDirectory() as tmp:
        train_path = Path(tmp) / "train.parquet"
        write_sequences(train_path)

        # column metadata drives decoding: scalar query_id, [MAX_LEN] item
        # sequence left-padded with 0 and a derived padding mask
        metadata = {"query_id": {"shape": []}, "item_id": {"shape": [MAX_LEN], "padding": 0}}
        dm = ParquetModule(
            metadata,
            batch_size=64,
            train_path=str(train_path),
            transforms=make_default_sasrec_transforms(tensor_schema),
            padding_mask_from="item_id",
        )

        model = SasRec.from_params(
            tensor_schema, max_sequence_length=MAX_LEN, embedding_dim=32, num_blocks=1
        )
        module = LightningModule(model, OptimizerFactory(lr=1e-3))
        trainer = Trainer(max_epochs=2, precision="bf16-mixed")
        trainer.fit(module, datamodule=dm)
        print("train_loss:", trainer.logged_metrics["train_loss"])

        model.eval()
        batch = {
            "item_id": torch.randint(0, N_ITEMS, (2, MAX_LEN)),
            "padding_mask": torch.ones(2, MAX_LEN, dtype=torch.bool),
        }
        print("top-5:", model.forward_inference(batch).topk(5, -1).indices)


if __name__ == "__main__":
    main()
