from .loggers import (
    CSVLogger,
    Logger,
    MLFlowLogger,
    TensorboardLogger,
    WandbLogger,
    generate_exp_name,
    get_logger,
)
from .recorder import Every, LoggerMonitor, PixelRenderTransform, TensorDictRecorder, VideoRecorder
