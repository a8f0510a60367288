from .common import (
    CSVLogger,
    Logger,
    MLFlowLogger,
    TensorboardLogger,
    WandbLogger,
    generate_exp_name,
    get_logger,
)
