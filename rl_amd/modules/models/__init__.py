from .models import (
    MLP,
    Conv3dNet,
    ConvNet,
    DdpgCnnActor,
    DdpgCnnQNet,
    DdpgMlpActor,
    DdpgMlpQNet,
    DuelingCnnDQNet,
    DuelingMlpDQNet,
    NormalParamExtractor,
)
