from .grpo import CISPOLoss, DAPO, DistillationLoss, GRPOLoss, MCAdvantage, SFTLoss
