from .actors import (
    Actor,
    ActorCriticOperator,
    ActorCriticWrapper,
    ActorValueOperator,
    DistributionalQValueActor,
    DistributionalQValueModule,
    MultiStepActorWrapper,
    ProbabilisticActor,
    QValueActor,
    QValueModule,
    TanhModule,
    ValueOperator,
)
from .exploration import (
    AdditiveGaussianModule,
    AdditiveGaussianWrapper,
    EGreedyModule,
    EGreedyWrapper,
    OrnsteinUhlenbeckProcessModule,
)
