from .evaluator import AlertEvaluator, AlertRule, AlertState, default_rules

__all__ = ["AlertEvaluator", "AlertRule", "AlertState", "default_rules"]
