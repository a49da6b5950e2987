"""Legacy logger surface (reference experimental/utils/logger.py) — thin
re-export of the main logging helper."""

from replay_amd.utils.session_handler import logger_with_settings

__all__ = ["logger_with_settings"]
