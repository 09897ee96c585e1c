from .plan_analyzer import PlanAnalyzer
from .candidate_analyzer import CandidateIndexAnalyzer
