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
from .loggers.common import ProcessLogger, RayLogger, TrackioLogger  # noqa: F401
__all__ = sorted(set(list(globals().get("__all__", [])) + ["ProcessLogger", "RayLogger"]))
