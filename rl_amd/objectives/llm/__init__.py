from .grpo import CISPOLoss, DAPO, GRPOLoss, MCAdvantage, SFTLoss
