from .replay_offline import OBP_AVAILABLE, OBPOfflinePolicyLearner, ips_estimate, snips_estimate

__all__ = ["OBP_AVAILABLE", "OBPOfflinePolicyLearner", "ips_estimate", "snips_estimate"]
