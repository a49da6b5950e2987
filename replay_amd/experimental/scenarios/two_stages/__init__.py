from .two_stages_scenario import TwoStagesScenario

__all__ = ["TwoStagesScenario"]
