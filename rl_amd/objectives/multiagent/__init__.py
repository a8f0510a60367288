from .mappo import IPPOLoss, MAPPOLoss
from .qmixer import QMixerLoss
