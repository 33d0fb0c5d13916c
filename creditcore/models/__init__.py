"""CPU reference models.

These define the exact numerics the MI355X HIP kernels are validated
against (fp32 GPU vs these fp64 CPU references):

- ``forest``   — sklearn classifier pipeline (reference 01-train cell-6)
- ``iforest``  — isolation-forest outlier detector with alibi-detect
                 ``IForest`` semantics (reference 02-register cell-6/9)
- ``drift``    — tabular drift detector with alibi-detect ``TabularDrift``
                 semantics: chi-square per categorical feature, two-sample
                 Kolmogorov-Smirnov per numeric feature (02-register cell-6)
- ``linear``   — logistic scorer (the "linear predict" model family)
"""

from .drift import TabularDriftDetector
from .iforest import IForestDetector
from .forest import make_classifier_pipeline

__all__ = ["TabularDriftDetector", "IForestDetector", "make_classifier_pipeline"]
