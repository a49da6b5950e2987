"""Model/encoder/splitter save-load registry.

Parity with reference replay/utils/model_handler.py (save:42, load:85,
registry lookup by class name :100, save_encoder:121, save_splitter:152) and
utils/common.py (save_to_replay:62, load_from_replay:71).
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Union


def save(model, path: Union[str, Path], overwrite: bool = False) -> None:
    base = Path(path)
    if base.exists() and not overwrite and any(base.iterdir()):
        raise FileExistsError(f"{base} already exists; pass overwrite=True")
    model.save(base)


def load(path: Union[str, Path], model_type=None):
    import replay_amd.models as models_module
    import replay_amd.scenarios as scenarios_module

    base = Path(path)
    meta = json.loads((base / "init_args.json").read_text())
    name = meta["_class_name"]
    klass = model_type
    if klass is None:
        klass = getattr(models_module, name, None) or getattr(scenarios_module, name, None)
    if klass is None:
        try:
            import replay_amd.experimental.models as exp_models

            klass = getattr(exp_models, name, None)
        except ImportError:  # pragma: no cover
            pass
    if klass is None:
        raise ValueError(f"Unknown model class {name}")
    init_args = meta["init_args"]
    # json stringifies None and bools under default=str; repair common cases
    cleaned = {}
    for key, value in init_args.items():
        if value == "None":
            cleaned[key] = None
        elif value == "True":
            cleaned[key] = True
        elif value == "False":
            cleaned[key] = False
        else:
            cleaned[key] = value
    model = klass(**cleaned)
    model._restore(base)
    return model


def save_encoder(encoder, path: Union[str, Path]) -> None:
    encoder.save(path)


def load_encoder(path: Union[str, Path]):
    from replay_amd.preprocessing import LabelEncoder

    return LabelEncoder.load(path)


def save_splitter(splitter, path: Union[str, Path]) -> None:
    splitter.save(path)


def load_splitter(path: Union[str, Path]):
    from replay_amd.splitters import Splitter

    return Splitter.load(path)


def save_to_replay(obj, path: Union[str, Path]) -> None:
    """Generic save for any replay_amd object exposing .save()."""
    obj.save(path)


def load_from_replay(path: Union[str, Path]):
    base = Path(path)
    if (base / "init_args.json").exists():
        meta = json.loads((base / "init_args.json").read_text())
        if "_class_name" in meta:
            name = meta["_class_name"]
            import replay_amd.splitters as splitters_module

            if hasattr(splitters_module, name):
                return load_splitter(base)
            return load(base)
    if (base / "label_encoder.pkl").exists():
        return load_encoder(base)
    raise ValueError(f"Cannot identify saved object at {path}")
