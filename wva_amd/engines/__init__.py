from .common import DecisionCache, DecisionTrigger, decision_to_optimized_alloc  # noqa: F401
