"""Cognitive subsystems: decay, temporal tracking, link prediction,
automatic relationship inference, Kalman filtering."""

from .decay import DecayConfig, DecayManager, HALF_LIVES
from .inference import InferenceConfig, InferenceEngine
from .kalman import Kalman1D
from .linkpredict import (PREDICTORS, adamic_adar, common_neighbors,
                          hybrid_score, jaccard, predict_links,
                          preferential_attachment, resource_allocation)
from .evidence import (CooldownTable, EvidenceBuffer, EvidenceThreshold)
from .patterns import (DetectedPattern, PatternConfig, PatternDetector,
                       QueryLoadPredictor)
from .temporal import AccessTracker, QueryLoadTracker, SESSION_GAP

__all__ = ["DecayManager", "DecayConfig", "HALF_LIVES", "InferenceEngine",
           "InferenceConfig", "Kalman1D", "AccessTracker", "QueryLoadTracker",
           "EvidenceBuffer", "EvidenceThreshold", "CooldownTable",
           "PatternDetector", "PatternConfig", "DetectedPattern",
           "QueryLoadPredictor",
           "SESSION_GAP", "predict_links", "common_neighbors", "jaccard",
           "adamic_adar", "preferential_attachment", "resource_allocation",
           "hybrid_score", "PREDICTORS"]
