"""Legacy save/load surface (reference experimental/utils/model_handler.py)
— re-exports the main model_handler; experimental models use the same
`.replay` directory format."""

from replay_amd.utils.model_handler import (
    load,
    load_encoder,
    load_splitter,
    save,
    save_encoder,
    save_splitter,
)

__all__ = ["load", "load_encoder", "load_splitter", "save", "save_encoder", "save_splitter"]
