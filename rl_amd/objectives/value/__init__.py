from .advantages import GAE, MultiAgentGAE, TD0Estimator, TD1Estimator, TDLambdaEstimator, ValueEstimatorBase, VTrace
from . import functional
