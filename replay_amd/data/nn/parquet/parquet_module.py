"""Data module over parquet splits.

Parity with reference ParquetModule (replay/data/nn/parquet/
parquet_module.py:19): per-split paths/batch sizes, transform pipelines
(torch.nn.Sequential of batch transforms) applied ON DEVICE in
``on_after_batch_transfer`` (:191), ``setup``:149,
``transfer_batch_to_device``:197.  Works with replay_amd.train.Trainer.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from .parquet_dataset import FixedBatchSizeDataset, ParquetDataset


class ParquetModule:
    def __init__(
        self,
        metadata: Dict[str, Dict],
        batch_size: int,
        train_path: Optional[str] = None,
        val_path: Optional[str] = None,
        test_path: Optional[str] = None,
        predict_path: Optional[str] = None,
        transforms: Optional[Dict[str, torch.nn.Sequential]] = None,
        num_workers: int = 0,
        shuffle_train: bool = True,
        seed: int = 0,
        padding_mask_from: Optional[str] = None,
    ) -> None:
        self.metadata = metadata
        self.batch_size = batch_size
        self.paths = {"train": train_path, "validate": val_path, "test": test_path, "predict": predict_path}
        self.transforms = transforms or {}
        self.num_workers = num_workers
        self.shuffle_train = shuffle_train
        self.seed = seed
        self.padding_mask_from = padding_mask_from
        self._datasets: Dict[str, FixedBatchSizeDataset] = {}

    def prepare_transforms(self, transforms: Dict[str, torch.nn.Sequential]) -> None:
        self.transforms = transforms

    def setup(self, stage: Optional[str] = None) -> None:
        stages = {"fit": ["train", "validate"], None: list(self.paths)}.get(stage, [stage])
        for s in stages:
            path = self.paths.get(s)
            if path is None or s in self._datasets:
                continue
            inner = ParquetDataset(
                path,
                self.batch_size,
                self.metadata,
                padding_mask_from=self.padding_mask_from,
                shuffle=(s == "train" and self.shuffle_train),
                seed=self.seed,
            )
            self._datasets[s] = FixedBatchSizeDataset(inner, self.batch_size)

    def _loader(self, split: str):
        if split not in self._datasets:
            self.setup(split)
        ds = self._datasets[split]
        return torch.utils.data.DataLoader(ds, batch_size=None, num_workers=self.num_workers)

    def train_dataloader(self):
        return self._loader("train")

    def val_dataloader(self):
        return self._loader("validate")

    def test_dataloader(self):
        return self._loader("test")

    def predict_dataloader(self):
        return self._loader("predict")

    # -- Trainer hooks ----------------------------------------------------------
    def transfer_batch_to_device(self, batch, device, dataloader_idx: int = 0):
        from replay_amd.train import move_batch

        return move_batch(batch, device)

    def on_after_batch_transfer(self, batch, stage: str):
        pipeline = self.transforms.get(stage)
        if pipeline is not None:
            batch = pipeline(batch)
        return batch
