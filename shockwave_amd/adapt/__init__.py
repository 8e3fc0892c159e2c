from .accordion import AccordionDetector, hardcoded_critical_regime
from .gns import GNSEstimator

__all__ = ["AccordionDetector", "hardcoded_critical_regime", "GNSEstimator"]
