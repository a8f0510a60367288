from .grpo import CISPOLoss, DAPO, DistillationLoss, GRPOLoss, MCAdvantage, SFTLoss
from .grpo import (
    CISPOLossOutput,
    DAPOLossOutput,
    DistillationLossOutput,
    GRPOLossOutput,
    LLMLossOutput,
    SFTLossOutput,
)
