from .continuous import (
    Delta,
    IndependentNormal,
    TanhDelta,
    TanhNormal,
    TruncatedNormal,
    safeatanh,
    safetanh,
)
from .discrete import LLMMaskedCategorical, MaskedCategorical, MaskedOneHotCategorical, OneHotCategorical, Ordinal
