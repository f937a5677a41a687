from .autoscaler import (Autoscaler, CronRecommender, DecayingHistogram,
                         ExternalRecommender, PercentileRecommender)

__all__ = ["Autoscaler", "PercentileRecommender", "CronRecommender",
           "ExternalRecommender", "DecayingHistogram"]
