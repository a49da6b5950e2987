"""Trainer checkpoint/resume + profiling-utils tests (aux subsystems)."""

import pytest
import torch

from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
from replay_amd.data.schema import FeatureHint, FeatureType
from replay_amd.nn.lightning import LightningModule, OptimizerFactory
from replay_amd.nn.sequential.sasrec import SasRec
from replay_amd.train import Trainer

pytestmark = pytest.mark.torch


def _loader(n=6):
    torch.manual_seed(0)
    batches = []
    for _ in range(n):
        b = {
            "item_id": torch.randint(0, 20, (4, 6)),
            "labels": torch.randint(0, 20, (4, 6)),
            "padding_mask": torch.ones(4, 6, dtype=torch.bool),
        }
        b["labels_padding_mask"] = b["padding_mask"]
        batches.append(b)
    return batches


def _module():
    schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id", FeatureType.CATEGORICAL, is_seq=True,
                feature_hint=FeatureHint.ITEM_ID, cardinality=20, embedding_dim=8,
            )
        ]
    )
    torch.manual_seed(1)
    model = SasRec.from_params(schema, max_sequence_length=6, embedding_dim=8, num_blocks=1, dropout=0.0)
    return LightningModule(model, OptimizerFactory(lr=1e-2))


def test_resume_from_checkpoint(tmp_path):
    loader = _loader()
    module = _module()
    trainer = Trainer(max_epochs=1, accelerator="cpu", precision="32")
    trainer.fit(module, loader)
    ckpt = tmp_path / "epoch1.ckpt"
    trainer.save_checkpoint(ckpt)

    # resume: epoch/global_step/optimizer state restored
    module2 = _module()
    trainer2 = Trainer(max_epochs=2, accelerator="cpu", precision="32")
    trainer2.fit(module2, loader, ckpt_path=str(ckpt))
    assert trainer2.current_epoch == 2  # completed epochs (Lightning convention)
    assert trainer2.global_step == 2 * len(loader)  # 6 before resume + 6 after
    # optimizer momentum actually restored (exp_avg nonzero from epoch 1)
    state = trainer2._optimizer.state_dict()["state"]
    assert len(state) > 0


def test_resume_weights_identical(tmp_path):
    loader = _loader()
    module = _module()
    trainer = Trainer(max_epochs=1, accelerator="cpu", precision="32")
    trainer.fit(module, loader)
    ckpt = tmp_path / "w.ckpt"
    trainer.save_checkpoint(ckpt)
    module2 = _module()
    Trainer(max_epochs=1, accelerator="cpu", precision="32")._prepare_eval(module2, str(ckpt))
    for p1, p2 in zip(module.parameters(), module2.parameters()):
        torch.testing.assert_close(p1, p2)


def test_roctx_range_noop_on_cpu():
    from replay_amd.utils.profiling import StepTimer, roctx_range

    with roctx_range("test"):
        pass
    timer = StepTimer()
    with timer.time("phase"):
        sum(range(1000))
    assert "phase" in timer.summary()


class TestTrainerCallbacks:
    def test_model_checkpoint_best_only(self, tmp_path):
        from replay_amd.train import ModelCheckpoint, Trainer

        module, loader = _module(), _loader()
        cb = ModelCheckpoint(dirpath=tmp_path, monitor="train_loss", mode="min")
        Trainer(max_epochs=3, accelerator="cpu", callbacks=[cb]).fit(module, loader)
        assert cb.best_model_path is not None
        import os

        assert os.path.exists(cb.best_model_path)
        assert cb.best_model_score is not None
        # only the single best checkpoint is kept
        assert len(list(tmp_path.glob("*.ckpt"))) == 1

    def test_early_stopping_stops(self):
        from replay_amd.train import EarlyStopping, Trainer

        module, loader = _module(), _loader()

        class ConstantMetric:
            def on_epoch_complete(self, trainer, mod):
                trainer.logged_metrics["plateau"] = 1.0

        es = EarlyStopping(monitor="plateau", patience=2, mode="min")
        tr = Trainer(max_epochs=50, accelerator="cpu", callbacks=[ConstantMetric(), es])
        tr.fit(module, loader)
        # first epoch sets best, then 2 epochs of no improvement stop it
        assert tr.current_epoch <= 4
        assert es.stopped_epoch is not None
