"""Objective utilities: ValueEstimators registry, target-net updaters,
distance losses, hold_out_net.

Reference: pytorch/rl torchrl/objectives/utils.py (ValueEstimators:48,
TargetNetUpdater:367, SoftUpdate:532, HardUpdate:591, hold_out_net:627).
"""
from __future__ import annotations

import contextlib
import enum
from typing import Iterator, Optional, Tuple, Union

import torch
from torch import nn

__all__ = [
    "ValueEstimators",
    "default_value_kwargs",
    "TargetNetUpdater",
    "SoftUpdate",
    "HardUpdate",
    "hold_out_net",
    "hold_out_params",
    "distance_loss",
    "next_state_value",
]


class ValueEstimators(enum.Enum):
    TD0 = "Bootstrapped TD (1-step return)"
    TD1 = "TD(1) (infinity-step return)"
    TDLambda = "TD(lambda)"
    GAE = "Generalized advantage estimate"
    VTrace = "V-trace"


def default_value_kwargs(value_type: ValueEstimators) -> dict:
    if value_type == ValueEstimators.TD0:
        return {"gamma": 0.99}
    if value_type == ValueEstimators.TD1:
        return {"gamma": 0.99}
    if value_type == ValueEstimators.TDLambda:
        return {"gamma": 0.99, "lmbda": 0.95}
    if value_type == ValueEstimators.GAE:
        return {"gamma": 0.99, "lmbda": 0.95}
    if value_type == ValueEstimators.VTrace:
        return {"gamma": 0.99}
    raise NotImplementedError(str(value_type))


def _target_pairs(loss_module) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
    """(source_param, target_buffer) pairs, matched by module path."""
    for name in loss_module._networks_with_targets():
        src = getattr(loss_module, name)
        tgt = getattr(loss_module, f"{name}_target")
        tgt_map = dict(tgt.named_buffers())
        for pname, p in src.named_parameters():
            t = tgt_map.get(pname)
            if t is not None:
                yield p, t
        # also sync source buffers (e.g. batch-norm stats)
        src_buf = dict(src.named_buffers())
        for pname, t in tgt_map.items():
            if pname in src_buf and src_buf[pname].shape == t.shape:
                s = src_buf[pname]
                if not s.requires_grad and pname not in dict(src.named_parameters()):
                    yield s, t


class TargetNetUpdater:
    """ABC for target-network sync (reference utils.py:367)."""

    def __init__(self, loss_module):
        self.loss_module = loss_module
        if not loss_module._networks_with_targets():
            raise RuntimeError(
                f"{type(loss_module).__name__} has no target networks to update"
            )
        for k in loss_module._has_update_associated:
            loss_module._has_update_associated[k] = True
        self.init_()

    def init_(self) -> None:
        with torch.no_grad():
            for p, t in _target_pairs(self.loss_module):
                t.copy_(p.detach())

    def step(self) -> None:
        with torch.no_grad():
            self._step()

    def _step(self) -> None:
        raise NotImplementedError


class SoftUpdate(TargetNetUpdater):
    """Polyak averaging: θ' ← (1-eps)·θ + eps·θ' … with
    eps = 1 - tau (reference utils.py:532)."""

    def __init__(self, loss_module, *, eps: Optional[float] = None, tau: Optional[float] = None):
        if eps is None and tau is None:
            eps = 0.999
        elif eps is None:
            eps = 1.0 - tau
        if not 0.0 <= eps <= 1.0:
            raise ValueError("eps must be in [0, 1]")
        self.eps = eps
        super().__init__(loss_module)

    def _step(self) -> None:
        eps = self.eps
        for p, t in _target_pairs(self.loss_module):
            if t.dtype.is_floating_point:
                t.mul_(eps).add_(p.detach(), alpha=1 - eps)
            else:
                t.copy_(p.detach())


class HardUpdate(TargetNetUpdater):
    """Copy every ``value_network_update_interval`` steps
    (reference utils.py:591)."""

    def __init__(self, loss_module, *, value_network_update_interval: int = 1000):
        self.value_network_update_interval = value_network_update_interval
        self.counter = 0
        super().__init__(loss_module)

    def _step(self) -> None:
        self.counter += 1
        if self.counter >= self.value_network_update_interval:
            self.counter = 0
            for p, t in _target_pairs(self.loss_module):
                t.copy_(p.detach())


@contextlib.contextmanager
def hold_out_net(net: nn.Module):
    """Temporarily disable grads on a network's params
    (reference utils.py:627)."""
    states = [p.requires_grad for p in net.parameters()]
    try:
        for p in net.parameters():
            p.requires_grad_(False)
        yield net
    finally:
        for p, s in zip(net.parameters(), states):
            p.requires_grad_(s)


hold_out_params = hold_out_net


def distance_loss(
    v1: torch.Tensor,
    v2: torch.Tensor,
    loss_function: str = "l2",
) -> torch.Tensor:
    """Pointwise distance (reference functional helpers)."""
    if loss_function == "l2":
        return torch.nn.functional.mse_loss(v1, v2, reduction="none")
    if loss_function == "l1":
        return torch.nn.functional.l1_loss(v1, v2, reduction="none")
    if loss_function in ("smooth_l1", "huber"):
        return torch.nn.functional.smooth_l1_loss(v1, v2, reduction="none")
    raise NotImplementedError(f"unknown loss_function {loss_function}")


def next_state_value(
    tensordict,
    operator=None,
    next_val_key: str = "state_action_value",
    gamma: float = 0.99,
    pred_next_val: Optional[torch.Tensor] = None,
    **kwargs,
) -> torch.Tensor:
    """r + γ·(1-done)·V' helper (legacy reference API)."""
    reward = tensordict.get(("next", "reward"))
    done = tensordict.get(("next", "done"))
    if pred_next_val is None:
        nxt = tensordict.get("next").clone(False)
        pred_next_val = operator(nxt).get(next_val_key)
    return reward + gamma * (~done).to(reward.dtype) * pred_next_val


RANDOM_MODULE_LIST: tuple = ()


def add_random_module(module) -> None:
    """Register a module class as stochastic so loss vmap-randomness
    detection treats it as random (reference common.py:1012)."""
    global RANDOM_MODULE_LIST
    RANDOM_MODULE_LIST = RANDOM_MODULE_LIST + (module,)


def group_optimizers(*optimizers) -> torch.optim.Optimizer:
    """Merge several same-type optimizers into one (reference
    utils.py:997): their param groups are concatenated so a single
    ``step()`` drives them all — one fused multi-tensor pass instead
    of several small ones."""
    cls = None
    groups = []
    for opt in optimizers:
        if opt is None:
            continue
        if cls is None:
            cls = type(opt)
        elif type(opt) is not cls:
            raise ValueError(
                f"all optimizers must share a type; got {cls.__name__} and {type(opt).__name__}"
            )
        groups.extend(opt.param_groups)
    if cls is None:
        raise ValueError("no optimizers given")
    merged = cls([{"params": []}])
    merged.param_groups = []
    for g in groups:
        merged.add_param_group(g)
    return merged


__all__ += ["add_random_module", "group_optimizers", "RANDOM_MODULE_LIST"]
