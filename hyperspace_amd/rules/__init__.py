from .apply_hyperspace import ApplyHyperspace
from .filter_reason import FilterReason, FilterReasons
from .candidate_collector import CandidateIndexCollector
from .score_optimizer import ScoreBasedIndexPlanOptimizer
