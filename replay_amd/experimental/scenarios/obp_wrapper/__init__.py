from replay_amd.utils import FeatureUnavailableError
from .replay_offline import OBP_AVAILABLE, OBPOfflinePolicyLearner, ips_estimate, snips_estimate

__all__ = [
    "FeatureUnavailableError","OBP_AVAILABLE", "OBPOfflinePolicyLearner", "ips_estimate", "snips_estimate"]
