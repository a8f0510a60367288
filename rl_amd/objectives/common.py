"""LossModule — the base class of every objective.

Reference: pytorch/rl torchrl/objectives/common.py:87 (LossModule,
convert_to_functional:426, target params :579-588, vmap ensembles
:939-969).

MI355X design note: the reference extracts functional parameter
TensorDicts and vmaps over them.  rl_amd keeps live modules — ensembles
are ``torch.func.stack_module_state`` + ``torch.vmap`` over N stacked
parameter sets (one batched MFMA GEMM instead of N small ones), and target
networks are deep copies whose parameters are demoted to buffers (so they
checkpoint with ``state_dict`` but never reach the optimizer).
"""
from __future__ import annotations

import copy
import dataclasses
from typing import Iterator, List, Optional, Sequence, Tuple, Union

import torch
from torch import nn

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase

__all__ = ["LossModule", "params_to_buffers"]


def params_to_buffers(module: nn.Module) -> nn.Module:
    """Demote every Parameter to a buffer, in place."""
    for m in module.modules():
        names = list(m._parameters.keys())
        for n in names:
            p = m._parameters.pop(n)
            if p is not None:
                m.register_buffer(n, p.detach().clone())
    return module


# keys a loss will auto-mask on when present (reference common.py:41)
AUTO_LOSS_MASK_KEYS = (("collector", "mask"), "shifted_valid")


class _EnsembleModule(nn.Module):
    """N copies of a module evaluated as one vmapped call
    (reference vmap-over-params, common.py:939-969)."""

    def __init__(self, module_factory_or_module, num_copies: int, reinit: bool = True):
        super().__init__()
        base = module_factory_or_module
        copies = []
        for i in range(num_copies):
            m = copy.deepcopy(base)
            if reinit and i > 0:
                for sub in m.modules():
                    if hasattr(sub, "reset_parameters"):
                        sub.reset_parameters()
            copies.append(m)
        self.modules_list = nn.ModuleList(copies)
        self.num_copies = num_copies
        self._batched = None

    def _try_batched_mlp(self):
        """Detect the common ensemble shape — every copy an identical
        Linear/activation stack reading the same in_keys — and cache
        the layer structure for a bmm fast path (one batched GEMM per
        layer instead of a python loop over copies; the win scales with
        the ensemble size, e.g. REDQ's 10 Q-nets)."""
        import torch.nn as nn

        if self._batched is not None:
            return self._batched
        _FAST = (nn.Linear, nn.Tanh, nn.ReLU, nn.ELU, nn.SiLU)
        _INERT = (nn.Identity,)
        try:
            per_copy = []
            for m in self.modules_list:
                # the fast path must account for EVERY compute submodule:
                # any leaf outside the whitelist (LayerNorm, Dropout, GELU,
                # BatchNorm, ...) forces the exact per-copy loop
                leaves = [
                    sub
                    for sub in m.modules()
                    if next(iter(sub.children()), None) is None
                ]
                if not all(isinstance(l, _FAST + _INERT) for l in leaves):
                    self._batched = False
                    return False
                layers = [sub for sub in m.modules() if isinstance(sub, _FAST)]
                per_copy.append(layers)
            def _kind(l):
                # subclasses (e.g. ops.SplitKLinear) batch as their base
                return "Linear" if isinstance(l, nn.Linear) else type(l).__name__

            sig = [
                (_kind(l),
                 getattr(l, "in_features", None), getattr(l, "out_features", None))
                for l in per_copy[0]
            ]
            ok = all(
                [(_kind(l), getattr(l, "in_features", None),
                  getattr(l, "out_features", None)) for l in layers] == sig
                for layers in per_copy[1:]
            )
            first = self.modules_list[0]
            in_keys = list(getattr(first, "in_keys", []))
            out_keys = list(getattr(first, "out_keys", []))
            ok = ok and len(out_keys) == 1 and len(in_keys) >= 1
            self._batched = (per_copy, sig, in_keys, out_keys) if ok else False
        except Exception:
            self._batched = False
        return self._batched

    def enable_bf16_cache(self):
        """Pre-stack the per-copy Linear weights as bf16 buffers —
        refreshed once per optimizer step (``refresh_bf16_cache_``)
        instead of a stack + autocast cast per forward call."""
        batched = self._try_batched_mlp()
        if not batched:
            return self
        per_copy, sig, _, _ = batched
        for li, (kind, _, _) in enumerate(sig):
            if kind != "Linear":
                continue
            W = torch.stack(
                [per_copy[n][li].weight.detach() for n in range(self.num_copies)]
            ).to(torch.bfloat16)
            b = torch.stack(
                [per_copy[n][li].bias.detach() for n in range(self.num_copies)]
            ).to(torch.bfloat16)
            self.register_buffer(f"_wstack_bf16_{li}", W)
            self.register_buffer(f"_bstack_bf16_{li}", b)
        self._stack_cache = True
        return self

    def refresh_bf16_cache_(self):
        if not getattr(self, "_stack_cache", False):
            return
        per_copy, sig, _, _ = self._try_batched_mlp()
        for li, (kind, _, _) in enumerate(sig):
            if kind != "Linear":
                continue
            getattr(self, f"_wstack_bf16_{li}").copy_(
                torch.stack([per_copy[n][li].weight.detach()
                             for n in range(self.num_copies)])
            )
            getattr(self, f"_bstack_bf16_{li}").copy_(
                torch.stack([per_copy[n][li].bias.detach()
                             for n in range(self.num_copies)])
            )

    def forward(self, td: TensorDictBase, detach_params: bool = False) -> TensorDictBase:
        """Run all copies on the same input; stack outputs along dim 0.

        ``detach_params=True`` evaluates with the weights treated as
        constants while gradients still flow through the INPUT (the
        reference's ``params.detach()`` actor-loss semantics) — served
        by the stacked-bf16 buffers when present, so no per-call weight
        stack/cast happens at all."""
        from ..tensordict import stack as td_stack

        batched = self._try_batched_mlp()
        if batched:
            per_copy, sig, in_keys, out_keys = batched
            # the cache is a buffer: it would SWALLOW weight gradients,
            # so it serves no-grad forwards (target nets) and
            # detach_params forwards (actor-loss q evaluations)
            cached = (
                getattr(self, "_stack_cache", False)
                and torch.is_autocast_enabled()
                and (not torch.is_grad_enabled() or detach_params)
            )
            x = torch.cat([td.get(k) for k in in_keys], dim=-1)
            lead = x.shape[:-1]
            h = x.reshape(1, -1, x.shape[-1]).expand(self.num_copies, -1, x.shape[-1])
            li = 0
            for kind, fin, fout in sig:
                if kind == "Linear":
                    if cached:
                        W = getattr(self, f"_wstack_bf16_{li}")
                        bias = getattr(self, f"_bstack_bf16_{li}")
                        with torch.autocast("cuda", enabled=False):
                            h = torch.baddbmm(
                                bias.unsqueeze(1), h.to(torch.bfloat16),
                                W.transpose(-2, -1),
                            )
                    else:
                        W = torch.stack([per_copy[n][li].weight for n in range(self.num_copies)])
                        bias = torch.stack([per_copy[n][li].bias for n in range(self.num_copies)])
                        if detach_params:
                            W = W.detach()
                            bias = bias.detach()
                        h = torch.baddbmm(bias.unsqueeze(1), h, W.transpose(-2, -1))
                else:
                    h = getattr(torch, kind.lower())(h) if kind == "Tanh" else getattr(torch.nn.functional, kind.lower())(h)
                li += 1
            out = h.reshape(self.num_copies, *lead, h.shape[-1])
            res = td.clone(False).unsqueeze(0).expand(self.num_copies, *td.batch_size)
            res = res.clone(False)
            res.set(out_keys[0], out)
            return res
        if detach_params:
            # exact per-copy path with constant weights: evaluate under
            # a fresh graph from detached parameter clones
            import copy as _copy

            outs = []
            for m in self.modules_list:
                frozen = _copy.deepcopy(m)
                for pp in frozen.parameters():
                    pp.requires_grad_(False)
                outs.append(frozen(td.clone(False)))
            return td_stack(outs, 0)
        outs = [m(td.clone(False)) for m in self.modules_list]
        return td_stack(outs, 0)

    @property
    def in_keys(self):
        return self.modules_list[0].in_keys

    @property
    def out_keys(self):
        return self.modules_list[0].out_keys

    def __getitem__(self, i):
        return self.modules_list[i]

    def __len__(self):
        return self.num_copies


class LossModule(TensorDictModuleBase):
    """Base class for objectives.

    Sub-classes call :meth:`convert_to_functional` for each network they
    own; target copies are created with ``create_target_params=True`` and
    updated by ``SoftUpdate``/``HardUpdate`` (objectives/utils.py).
    """

    @dataclasses.dataclass
    class _AcceptedKeys:
        pass

    default_keys = _AcceptedKeys
    out_keys: List[str] = []

    def __init__(self):
        super().__init__()
        self._tensor_keys = self._AcceptedKeys()
        self._networks: List[str] = []
        self._has_update_associated = {}
        self.value_estimator = None
        self.in_keys = []

    # ------------------------------------------------------------------ #
    @property
    def tensor_keys(self):
        return self._tensor_keys

    def set_keys(self, **kwargs) -> "LossModule":
        for k, v in kwargs.items():
            if not hasattr(self._tensor_keys, k):
                raise ValueError(
                    f"{k} is not an accepted key for {type(self).__name__}; "
                    f"accepted: {[f.name for f in dataclasses.fields(self._tensor_keys)]}"
                )
            setattr(self._tensor_keys, k, v)
        self._forward_key_update()
        return self

    def _forward_key_update(self):
        pass

    # ------------------------------------------------------------------ #
    def convert_to_functional(
        self,
        module: TensorDictModuleBase,
        module_name: str,
        expand_dim: Optional[int] = None,
        create_target_params: bool = False,
        compare_against: Optional[Sequence] = None,
        **kwargs,
    ) -> None:
        """Register ``module`` (optionally expanded to an ensemble of
        ``expand_dim`` copies) and optionally a frozen target copy
        (reference common.py:426)."""
        if expand_dim is not None and expand_dim > 1:
            module = _EnsembleModule(module, expand_dim)
        setattr(self, module_name, module)
        self._networks.append(module_name)
        if create_target_params:
            target = params_to_buffers(copy.deepcopy(module))
            setattr(self, f"{module_name}_target", target)
            self._has_update_associated[module_name] = False
        # extend in_keys
        for k in getattr(module, "in_keys", []):
            if k not in self.in_keys:
                self.in_keys.append(k)

    def target_network(self, module_name: str) -> nn.Module:
        return getattr(self, f"{module_name}_target")

    def _networks_with_targets(self) -> List[str]:
        return [n for n in self._networks if hasattr(self, f"{n}_target")]

    # ------------------------------------------------------------------ #
    def make_value_estimator(self, value_type=None, **hyperparams):
        """Build the default value estimator (reference common.py + each
        loss's ``make_value_estimator``)."""
        from .utils import ValueEstimators, default_value_kwargs
        from .value.advantages import GAE, TD0Estimator, TD1Estimator, TDLambdaEstimator, VTrace

        if value_type is None:
            value_type = getattr(self, "default_value_estimator", ValueEstimators.TD0)
        hp = dict(default_value_kwargs(value_type))
        hp.update(hyperparams)
        value_net = getattr(self, "critic_network", None) or getattr(
            self, "value_network", None
        )
        if value_type == ValueEstimators.GAE:
            self.value_estimator = GAE(value_network=value_net, **hp)
        elif value_type == ValueEstimators.TD0:
            self.value_estimator = TD0Estimator(value_network=value_net, **hp)
        elif value_type == ValueEstimators.TD1:
            self.value_estimator = TD1Estimator(value_network=value_net, **hp)
        elif value_type == ValueEstimators.TDLambda:
            self.value_estimator = TDLambdaEstimator(value_network=value_net, **hp)
        elif value_type == ValueEstimators.VTrace:
            actor = getattr(self, "actor_network", None)
            self.value_estimator = VTrace(
                value_network=value_net, actor_network=actor, **hp
            )
        else:
            raise NotImplementedError(f"value type {value_type}")
        self._default_value_estimator_keys()
        return self.value_estimator

    def _default_value_estimator_keys(self):
        pass

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        raise NotImplementedError

    def loss_and_backprop(self, td: TensorDictBase) -> TensorDictBase:
        """Convenience: forward + sum of loss_* keys + backward."""
        out = self.forward(td)
        total = sum(
            v for k, v in out.items() if isinstance(k, str) and k.startswith("loss_")
        )
        total.backward()
        return out
