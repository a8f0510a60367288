from .mocking_classes import (
    ContinuousActionVecMockEnv,
    CountingEnv,
    DiscreteActionVecMockEnv,
    EnvThatErrors,
    EnvWithDynamicSpec,
    HeterogeneousCountingEnv,
    MockSerialEnv,
    MultiKeyCountingEnv,
    NestedCountingEnv,
)
from .dist_utils import assert_no_new_python_processes, snapshot_python_processes
