from .evaluator import (AlertEvaluator, AlertRule, AlertState, default_rules,
                        rules_from_config)

__all__ = ["AlertEvaluator", "AlertRule", "AlertState", "default_rules",
           "rules_from_config"]
