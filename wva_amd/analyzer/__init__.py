"""Queueing analytics for LLM inference servers (layer L5).

Re-implements, from scratch, the behavior of the reference analytics library
(/root/reference/pkg/analyzer: queueanalyzer.go, mm1modelstatedependent.go,
mm1kmodel.go, queuemodel.go, utils.go) with better numerics (log-space
product-form probabilities instead of MaxFloat rescale loops) and no
package-global evaluation state (closures capture the model, so the library
is thread-safe by construction).
"""

from .queuemodel import QueueModel, MM1KModel
from .statedep import MM1ModelStateDependent
from .mg1 import MG1Corrector, MG1Metrics, auto_scv_enabled, configured_scv, pollaczek_khinchine_wait, recommended_service_scv, service_scv_from_tokens
from .search import BelowRegion, InRegion, AboveRegion, binary_search, within_tolerance
from .queueanalyzer import (
    EPSILON,
    STABILITY_SAFETY_FRACTION,
    AnalyzerError,
    AnalysisMetrics,
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RateRange,
    RequestSize,
    ServiceParms,
    TargetPerf,
    TargetRate,
    effective_concurrency,
)

__all__ = [
    "QueueModel",
    "MM1KModel",
    "MM1ModelStateDependent",
    "BelowRegion",
    "InRegion",
    "AboveRegion",
    "binary_search",
    "within_tolerance",
    "EPSILON",
    "STABILITY_SAFETY_FRACTION",
    "AnalyzerError",
    "AnalysisMetrics",
    "Configuration",
    "DecodeParms",
    "PrefillParms",
    "QueueAnalyzer",
    "RateRange",
    "RequestSize",
    "ServiceParms",
    "TargetPerf",
    "TargetRate",
    "effective_concurrency",
    "MG1Corrector",
    "configured_scv",
    "auto_scv_enabled",
    "recommended_service_scv",
    "MG1Metrics",
    "pollaczek_khinchine_wait",
    "service_scv_from_tokens",
]
