from .mocking_classes import (
    ContinuousActionVecMockEnv,
    CountingEnv,
    DiscreteActionVecMockEnv,
    EnvThatErrors,
    MockSerialEnv,
    NestedCountingEnv,
)
