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
from .rnn import (
    GRUCell,
    GRUModule,
    LSTMCell,
    LSTMModule,
    gru_scan,
    lstm_scan,
    recurrent_mode,
    set_recurrent_mode,
)
from .actors import DecisionTransformerInferenceWrapper, LMHeadActorValueOperator
from .common import SafeModule, SafeProbabilisticModule, SafeSequential
