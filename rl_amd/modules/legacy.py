"""Legacy/alias API names kept for reference parity.

Reference: pytorch/rl keeps deprecated wrapper forms alongside the
module forms (tensordict_module/exploration.py EGreedyWrapper etc.);
rl_amd exposes them as thin aliases/subclasses of the module forms.
"""
from __future__ import annotations

import torch
from torch import nn

from .distributions import TanhNormal  # noqa: F401  (re-export convenience)
from .models.models import NormalParamExtractor
from .tensordict_module.exploration import (
    AdditiveGaussianModule,
    EGreedyModule,
    OrnsteinUhlenbeckProcessModule,
)

__all__ = [
    "EGreedyWrapper",
    "AdditiveGaussianWrapper",
    "OrnsteinUhlenbeckProcessWrapper",
    "NormalParamWrapper",
    "QValueHook",
    "DistributionalQValueHook",
    "WorldModel",
    "GRU",
    "LSTM",
    "GRUBase",
    "LSTMBase",
]


def _wrapperize(module_cls):
    """Deprecated wrapper form: wraps a policy and appends the
    exploration module behind it (the module form composes with
    TensorDictSequential instead)."""

    class _Wrapper(nn.Module):
        def __init__(self, policy, *args, **kwargs):
            super().__init__()
            self.policy = policy
            self.exploration = module_cls(*args, **kwargs)
            self.in_keys = getattr(policy, "in_keys", [])
            self.out_keys = getattr(policy, "out_keys", [])

        def forward(self, td):
            return self.exploration(self.policy(td))

        def step(self, n: int = 1):
            if hasattr(self.exploration, "step"):
                self.exploration.step(n)

    _Wrapper.__name__ = module_cls.__name__.replace("Module", "Wrapper")
    return _Wrapper


EGreedyWrapper = _wrapperize(EGreedyModule)
AdditiveGaussianWrapper = _wrapperize(AdditiveGaussianModule)
OrnsteinUhlenbeckProcessWrapper = _wrapperize(OrnsteinUhlenbeckProcessModule)


class NormalParamWrapper(nn.Module):
    """Deprecated form of :class:`NormalParamExtractor` wrapping the
    producing network (reference distributions/utils.py)."""

    def __init__(self, operator: nn.Module, scale_mapping: str = "biased_softplus_1.0", scale_lb: float = 1e-4):
        super().__init__()
        self.operator = operator
        self.extractor = NormalParamExtractor(scale_mapping=scale_mapping, scale_lb=scale_lb)

    def forward(self, *args, **kwargs):
        out = self.operator(*args, **kwargs)
        return self.extractor(out)


class QValueHook:
    """Legacy hook form of QValueModule (reference actors.py QValueHook):
    call on (net output) -> (action, values, chosen value)."""

    def __init__(self, action_space: str = "one_hot"):
        self.action_space = action_space

    def __call__(self, values: torch.Tensor):
        if self.action_space == "categorical":
            action = values.argmax(-1)
            chosen = values.gather(-1, action.unsqueeze(-1)).squeeze(-1)
        else:
            idx = values.argmax(-1, keepdim=True)
            action = torch.zeros_like(values).scatter_(-1, idx, 1.0)
            chosen = values.gather(-1, idx).squeeze(-1)
        return action, values, chosen


class DistributionalQValueHook(QValueHook):
    """Legacy hook over distributional logits [..., atoms, actions]."""

    def __init__(self, action_space: str = "one_hot", support: torch.Tensor = None):
        super().__init__(action_space)
        self.support = support

    def __call__(self, log_probs: torch.Tensor):
        probs = log_probs.exp()
        q = (probs * self.support.unsqueeze(-1)).sum(-2)
        return super().__call__(q)


# world-model alias + recurrent re-exports (reference models.py WorldModel,
# tensordict_module/rnn.py GRU/LSTM re-exports of the torch cells with
# TensorDict glue living in GRUModule/LSTMModule)
from .models.model_based import WorldModelWrapper as WorldModel  # noqa: E402

GRU = nn.GRU
LSTM = nn.LSTM
GRUBase = nn.GRU
LSTMBase = nn.LSTM
