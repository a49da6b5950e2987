from .two_stages.two_stages_scenario import TwoStagesScenario

__all__ = ["TwoStagesScenario"]
