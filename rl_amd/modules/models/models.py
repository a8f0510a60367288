"""Config-driven network builders: MLP, ConvNet, dueling heads, DDPG nets.

Reference: pytorch/rl torchrl/modules/models/models.py:29 (MLP), :305
(ConvNet), :819-1401 (dueling / DDPG heads).  The actor-MLP forward is a
HIP-fusion target (see rl_amd/ops): at inference the MLP chain + TanhNormal
epilogue runs as one fused CDNA4 kernel when shapes allow.
"""
from __future__ import annotations

from numbers import Number
from typing import Callable, List, Optional, Sequence, Type, Union

import torch
from torch import nn

__all__ = [
    "MLP",
    "ConvNet",
    "Conv3dNet",
    "DuelingMlpDQNet",
    "DuelingCnnDQNet",
    "DdpgMlpActor",
    "DdpgMlpQNet",
    "DdpgCnnActor",
    "DdpgCnnQNet",
    "NormalParamExtractor",
]


class SquashDims(nn.Module):
    def __init__(self, ndims_in: int = 3):
        super().__init__()
        self.ndims_in = ndims_in

    def forward(self, x):
        return x.flatten(-self.ndims_in)


def _make_activation(activation_class, **kwargs):
    return activation_class(**kwargs)


class MLP(nn.Sequential):
    """Multi-layer perceptron with config-driven depth/width/norm/activation
    (reference models.py:29).  ``num_cells`` int → repeated ``depth`` times,
    or explicit list."""

    def __init__(
        self,
        in_features: Optional[int] = None,
        out_features: int = None,
        depth: Optional[int] = None,
        num_cells: Union[int, Sequence[int], None] = None,
        activation_class: Type[nn.Module] = nn.Tanh,
        activation_kwargs: Optional[dict] = None,
        norm_class: Optional[Type[nn.Module]] = None,
        norm_kwargs: Optional[dict] = None,
        bias_last_layer: bool = True,
        single_bias_last_layer: bool = False,
        layer_class: Type[nn.Module] = nn.Linear,
        layer_kwargs: Optional[dict] = None,
        activate_last_layer: bool = False,
        device=None,
        dtype=None,
    ):
        if out_features is None:
            raise ValueError("out_features must be specified")
        if num_cells is None:
            num_cells = [32, 32, 32] if depth is None else [32] * depth
        if isinstance(num_cells, Number):
            num_cells = [int(num_cells)] * (depth if depth is not None else 3)
        num_cells = list(num_cells)
        self.in_features = in_features
        self.out_features = out_features
        self.num_cells = num_cells
        activation_kwargs = activation_kwargs or {}
        norm_kwargs = norm_kwargs or {}
        layer_kwargs = layer_kwargs or {}
        factory = {"device": device, "dtype": dtype}

        layers: List[nn.Module] = []
        dims_in = [in_features, *num_cells]
        dims_out = [*num_cells, out_features]
        n = len(dims_out)
        for i, (din, dout) in enumerate(zip(dims_in, dims_out)):
            last = i == n - 1
            if din is None:
                layers.append(nn.LazyLinear(dout, bias=bias_last_layer if last else True, **factory))
            else:
                layers.append(
                    layer_class(din, dout, bias=bias_last_layer if last else True, **layer_kwargs, **factory)
                )
            if not last or activate_last_layer:
                if norm_class is not None:
                    layers.append(norm_class(dout, **norm_kwargs))
                layers.append(_make_activation(activation_class, **activation_kwargs))
        super().__init__(*layers)

    def forward(self, *inputs):
        if len(inputs) > 1:
            inputs = (torch.cat([i for i in inputs], dim=-1),)
        x = inputs[0]
        # flatten leading batch dims beyond 1 for Linear efficiency
        lead = x.shape[:-1]
        if len(lead) > 1:
            out = super().forward(x.reshape(-1, x.shape[-1]))
            return out.reshape(*lead, out.shape[-1])
        return super().forward(x)


class ConvNet(nn.Sequential):
    """2-D conv stack + flatten (reference models.py:305)."""

    def __init__(
        self,
        in_features: Optional[int] = None,
        depth: Optional[int] = None,
        num_cells: Union[int, Sequence[int], None] = None,
        kernel_sizes: Union[int, Sequence[int]] = 3,
        strides: Union[int, Sequence[int]] = 1,
        paddings: Union[int, Sequence[int]] = 0,
        activation_class: Type[nn.Module] = nn.ELU,
        activation_kwargs: Optional[dict] = None,
        norm_class: Optional[Type[nn.Module]] = None,
        norm_kwargs: Optional[dict] = None,
        aggregator_class: Optional[Type[nn.Module]] = SquashDims,
        aggregator_kwargs: Optional[dict] = None,
        squeeze_output: bool = False,
        device=None,
        dtype=None,
    ):
        if num_cells is None:
            num_cells = [32, 32, 32] if depth is None else [32] * depth
        if isinstance(num_cells, Number):
            num_cells = [int(num_cells)] * (depth if depth is not None else 3)
        num_cells = list(num_cells)
        n = len(num_cells)

        def _bcast(v):
            if isinstance(v, Number):
                return [v] * n
            return list(v)

        kernel_sizes = _bcast(kernel_sizes)
        strides = _bcast(strides)
        paddings = _bcast(paddings)
        activation_kwargs = activation_kwargs or {}
        norm_kwargs = norm_kwargs or {}
        factory = {"device": device, "dtype": dtype}
        layers: List[nn.Module] = []
        dims_in = [in_features, *num_cells[:-1]]
        for i, (din, dout) in enumerate(zip(dims_in, num_cells)):
            if din is None:
                layers.append(
                    nn.LazyConv2d(dout, kernel_sizes[i], strides[i], paddings[i], **factory)
                )
            else:
                layers.append(
                    nn.Conv2d(din, dout, kernel_sizes[i], strides[i], paddings[i], **factory)
                )
            if norm_class is not None:
                layers.append(norm_class(dout, **norm_kwargs))
            layers.append(_make_activation(activation_class, **activation_kwargs))
        if aggregator_class is not None:
            layers.append(aggregator_class(**(aggregator_kwargs or {"ndims_in": 3})))
        super().__init__(*layers)

    def forward(self, x):
        lead = x.shape[:-3]
        if len(lead) > 1:
            out = super().forward(x.reshape(-1, *x.shape[-3:]))
            return out.reshape(*lead, *out.shape[1:])
        if len(lead) == 0:
            return super().forward(x.unsqueeze(0)).squeeze(0)
        return super().forward(x)


class Conv3dNet(ConvNet):
    """3-D conv variant (reference models.py:572) — shares ConvNet config."""

    def __init__(self, *args, **kwargs):
        kwargs.setdefault("aggregator_kwargs", {"ndims_in": 4})
        super().__init__(*args, **kwargs)
        # swap Conv2d for Conv3d
        for i, m in enumerate(self):
            if isinstance(m, nn.Conv2d):
                new = nn.Conv3d(
                    m.in_channels, m.out_channels, m.kernel_size[0], m.stride[0], m.padding[0]
                )
                self[i] = new
            elif isinstance(m, nn.LazyConv2d):
                self[i] = nn.LazyConv3d(
                    m.out_channels, m.kernel_size[0], m.stride[0], m.padding[0]
                )

    def forward(self, x):
        lead = x.shape[:-4]
        if len(lead) > 1:
            out = nn.Sequential.forward(self, x.reshape(-1, *x.shape[-4:]))
            return out.reshape(*lead, *out.shape[1:])
        if len(lead) == 0:
            return nn.Sequential.forward(self, x.unsqueeze(0)).squeeze(0)
        return nn.Sequential.forward(self, x)


class DuelingMlpDQNet(nn.Module):
    """Dueling Q-net: value + advantage streams (reference models.py:819)."""

    def __init__(
        self,
        out_features: int,
        out_features_value: int = 1,
        mlp_kwargs_feature: Optional[dict] = None,
        mlp_kwargs_output: Optional[dict] = None,
        device=None,
    ):
        super().__init__()
        mlp_kwargs_feature = mlp_kwargs_feature or {
            "num_cells": [128, 128],
            "out_features": 128,
            "activate_last_layer": True,
        }
        self.features = MLP(device=device, **mlp_kwargs_feature)
        mlp_kwargs_output = mlp_kwargs_output or {"num_cells": [64], "depth": 1}
        self.advantage = MLP(out_features=out_features, device=device, **{k: v for k, v in mlp_kwargs_output.items() if k != "out_features"})
        self.value = MLP(out_features=out_features_value, device=device, **{k: v for k, v in mlp_kwargs_output.items() if k != "out_features"})

    def forward(self, x):
        h = self.features(x)
        adv = self.advantage(h)
        val = self.value(h)
        return val + adv - adv.mean(-1, keepdim=True)


class DuelingCnnDQNet(nn.Module):
    """CNN dueling Q-net (reference models.py:936)."""

    def __init__(
        self,
        out_features: int,
        out_features_value: int = 1,
        cnn_kwargs: Optional[dict] = None,
        mlp_kwargs: Optional[dict] = None,
        device=None,
    ):
        super().__init__()
        cnn_kwargs = cnn_kwargs or {
            "num_cells": [32, 64, 64],
            "strides": [4, 2, 1],
            "kernel_sizes": [8, 4, 3],
        }
        self.features = ConvNet(device=device, **cnn_kwargs)
        mlp_kwargs = mlp_kwargs or {"num_cells": [512]}
        self.advantage = MLP(out_features=out_features, device=device, **mlp_kwargs)
        self.value = MLP(out_features=out_features_value, device=device, **mlp_kwargs)

    def forward(self, x):
        h = self.features(x)
        adv = self.advantage(h)
        val = self.value(h)
        return val + adv - adv.mean(-1, keepdim=True)


class DistributionalDQNnet(nn.Module):
    """Log-softmax head for distributional (C51-style) Q-networks
    (reference models.py DistributionalDQNnet): normalizes the atom
    dimension of a ``[*, n_atoms, n_actions]`` logit tensor so
    :class:`~rl_amd.modules.DistributionalQValueModule` receives proper
    log-probabilities."""

    def __init__(self, in_keys=None, out_keys=None):
        super().__init__()
        self.in_keys = in_keys
        self.out_keys = out_keys

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() < 2:
            raise ValueError("DistributionalDQNnet expects [*, n_atoms, n_actions]")
        return torch.log_softmax(x, dim=-2)


class DdpgMlpActor(nn.Module):
    """DDPG MLP actor (reference models.py:1207)."""

    def __init__(self, action_dim: int, mlp_net_kwargs: Optional[dict] = None, device=None):
        super().__init__()
        kwargs = mlp_net_kwargs or {"num_cells": [400, 300], "activation_class": nn.ReLU}
        self.mlp = MLP(out_features=action_dim, device=device, **kwargs)

    def forward(self, obs):
        return self.mlp(obs)


class DdpgMlpQNet(nn.Module):
    """DDPG MLP Q-net: obs through net1, [h, action] through net2
    (reference models.py:1401)."""

    def __init__(self, mlp_net_kwargs_net1: Optional[dict] = None, mlp_net_kwargs_net2: Optional[dict] = None, device=None):
        super().__init__()
        k1 = mlp_net_kwargs_net1 or {
            "num_cells": [400],
            "out_features": 400,
            "activation_class": nn.ReLU,
            "activate_last_layer": True,
        }
        k2 = mlp_net_kwargs_net2 or {
            "num_cells": [300],
            "out_features": 1,
            "activation_class": nn.ReLU,
        }
        self.mlp1 = MLP(device=device, **k1)
        self.mlp2 = MLP(device=device, **k2)

    def forward(self, obs, action):
        h = self.mlp1(obs)
        return self.mlp2(torch.cat([h, action], -1))


class DdpgCnnActor(nn.Module):
    """DDPG CNN actor (reference models.py:1081)."""

    def __init__(self, action_dim: int, conv_net_kwargs: Optional[dict] = None, mlp_net_kwargs: Optional[dict] = None, device=None):
        super().__init__()
        self.convnet = ConvNet(device=device, **(conv_net_kwargs or {}))
        self.mlp = MLP(
            out_features=action_dim,
            device=device,
            **(mlp_net_kwargs or {"num_cells": [200, 200]}),
        )

    def forward(self, obs):
        return self.mlp(self.convnet(obs))


class DdpgCnnQNet(nn.Module):
    """DDPG CNN Q-net (reference models.py:1278)."""

    def __init__(self, conv_net_kwargs: Optional[dict] = None, mlp_net_kwargs: Optional[dict] = None, device=None):
        super().__init__()
        self.convnet = ConvNet(device=device, **(conv_net_kwargs or {}))
        self.mlp = MLP(
            out_features=1, device=device, **(mlp_net_kwargs or {"num_cells": [200, 200]})
        )

    def forward(self, obs, action):
        h = self.convnet(obs)
        return self.mlp(torch.cat([h, action], -1))


class NormalParamExtractor(nn.Module):
    """Split trailing dim into (loc, scale) with positive-mapped scale
    (reference: tensordict.nn.NormalParamExtractor, used throughout
    pytorch/rl actor construction)."""

    def __init__(self, scale_mapping: str = "biased_softplus_1.0", scale_lb: float = 1e-4):
        super().__init__()
        self.scale_mapping = scale_mapping
        self.scale_lb = scale_lb
        if scale_mapping.startswith("biased_softplus"):
            bias = float(scale_mapping.rsplit("_", 1)[-1]) if "_" in scale_mapping[len("biased_softplus"):] else 1.0
            import math

            self._inv_softplus_bias = math.log(math.expm1(bias))
        else:
            self._inv_softplus_bias = None

    def forward(self, x):
        loc, scale = x.chunk(2, -1)
        if self.scale_mapping == "exp":
            scale = scale.clamp(-20.0, 2.0).exp()
        elif self._inv_softplus_bias is not None:
            scale = torch.nn.functional.softplus(scale + self._inv_softplus_bias)
        else:  # softplus
            scale = torch.nn.functional.softplus(scale)
        return loc, scale.clamp_min(self.scale_lb)
