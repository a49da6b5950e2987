"""Import harness for the READ-ONLY reference checkout at /root/reference.

Used by quality-parity tests only: it registers minimal fake ``polars`` and
``lightning`` modules (any missing attribute resolves to a permissive dummy)
so the reference's raw torch modules import in this Spark/Lightning-free
environment.  Nothing from the reference is copied — its code is executed
in place purely as a numerical oracle.
"""

from __future__ import annotations

import importlib.machinery
import sys
import types

REFERENCE_ROOT = "/root/reference"


class _Dummy:
    """Permissive stand-in: constructible, callable, subscriptable."""

    def __init__(self, *a, **k):
        pass

    def __call__(self, *a, **k):
        return self

    def __class_getitem__(cls, item):
        return cls

    def __getattr__(self, name):
        return _Dummy()


def _fake_module(name: str, attrs=None):
    import torch

    mod = types.ModuleType(name)
    mod.__spec__ = importlib.machinery.ModuleSpec(name, None)
    mod.__path__ = []

    base_attrs = {
        "LightningModule": torch.nn.Module,
        "LightningDataModule": object,
        "Trainer": _Dummy,
        "Callback": _Dummy,
    }
    base_attrs.update(attrs or {})
    for key, value in base_attrs.items():
        setattr(mod, key, value)

    def module_getattr(name_, _mod=mod):  # PEP 562 fallback
        if name_.startswith("__"):  # keep inspect/importlib sane
            raise AttributeError(name_)
        return _Dummy
    mod.__getattr__ = module_getattr
    sys.modules[name] = mod
    return mod


def install_fakes() -> None:
    if "polars" in sys.modules and getattr(sys.modules["polars"], "__fake__", False):
        return
    pl = _fake_module("polars")
    pl.__fake__ = True
    for sub in [
        "lightning",
        "lightning.pytorch",
        "lightning.pytorch.trainer",
        "lightning.pytorch.trainer.states",
        "lightning.pytorch.utilities",
        "lightning.pytorch.callbacks",
        "lightning.pytorch.loggers",
        "lightning.pytorch.strategies",
        "lightning.fabric",
        "lightning.fabric.utilities",
    ]:
        _fake_module(sub)

    class _RunningStage:
        TRAINING = "train"
        VALIDATING = "validate"
        TESTING = "test"
        PREDICTING = "predict"
        SANITY_CHECKING = "sanity"

    sys.modules["lightning.pytorch.trainer.states"].RunningStage = _RunningStage
    sys.modules["lightning.pytorch.utilities"].CombinedLoader = _Dummy
    import torch

    sys.modules["lightning.pytorch.utilities"].move_data_to_device = (
        lambda batch, device: batch
    )


def load_reference():
    """Returns the imported ``replay`` package from /root/reference."""
    install_fakes()
    if REFERENCE_ROOT not in sys.path:
        sys.path.insert(0, REFERENCE_ROOT)
    import replay  # noqa: F401

    return replay
