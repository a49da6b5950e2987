"""Process-level state + logging.

The reference's session handler manages a JVM Spark session singleton
(replay/utils/session_handler.py:22,130) and the "replay" logger
(:106-119).  The MI355X build has no JVM: State holds the torch device and
seeds; get_session is the compatibility entry point.
"""

from __future__ import annotations

import logging
from typing import Optional


class Borg:
    _shared_state: dict = {}

    def __init__(self):
        self.__dict__ = self._shared_state


class State(Borg):
    """Process-level singleton: device + RNG seed (the reference's Spark
    session slot, repurposed for the GPU runtime)."""

    def __init__(self, device: Optional[str] = None, seed: Optional[int] = None):
        super().__init__()
        if device is not None or not hasattr(self, "device"):
            import torch

            if device is None:
                device = "cuda" if torch.cuda.is_available() else "cpu"
            self.device = torch.device(device)
        if seed is not None:
            import torch

            self.seed = seed
            torch.manual_seed(seed)


def get_session(device: Optional[str] = None) -> State:
    return State(device)


def logger_with_settings(level: int = logging.INFO) -> logging.Logger:
    """Configure and return the 'replay' logger (reference :106-119)."""
    logger = logging.getLogger("replay_amd")
    logger.setLevel(level)
    if not logger.handlers:
        handler = logging.StreamHandler()
        handler.setFormatter(logging.Formatter("%(asctime)s %(levelname)s %(name)s: %(message)s"))
        logger.addHandler(handler)
    return logger


def get_spark_session(*args, **kwargs):
    raise RuntimeError(
        "Spark is not part of the MI355X build of replay_amd: classical models "
        "run pandas/numpy-native and NN models run on ROCm GPUs. "
        "Use replay_amd.utils.session_handler.get_session() instead."
    )
