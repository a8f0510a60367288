from .continuous import (
    Delta,
    IndependentNormal,
    TanhDelta,
    TanhNormal,
    TruncatedNormal,
    safeatanh,
    safetanh,
)
from .discrete import MaskedCategorical, MaskedOneHotCategorical, OneHotCategorical, Ordinal
