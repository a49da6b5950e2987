"""Legacy session surface (reference experimental/utils/session_handler.py)
— the State singleton replaces the Spark session registry."""

from replay_amd.utils.session_handler import State, logger_with_settings

__all__ = ["State", "logger_with_settings"]
