from .advantages import GAE, TD0Estimator, TD1Estimator, TDLambdaEstimator, ValueEstimatorBase, VTrace
from . import functional
