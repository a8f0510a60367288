"""TensorDict-keyed maps and MCTS forest.

Reference: pytorch/rl torchrl/data/map/ (SipHash/RandomProjectionHash
hash.py:75,119, QueryModule query.py:59, TensorDictMap tdstorage.py:59,
Tree/MCTSForest tree.py:30,682) and torchrl/modules/mcts/scores.py
(PUCT:34, UCB:150).
"""
from __future__ import annotations

import hashlib
from typing import Any, Callable, Dict, List, Optional, Sequence

import torch

from ..tensordict import TensorDict, TensorDictBase, stack as td_stack

__all__ = [
    "SipHash",
    "RandomProjectionHash",
    "QueryModule",
    "TensorDictMap",
    "Tree",
    "MCTSForest",
    "PUCTScore",
    "UCBScore",
]


class SipHash:
    """Deterministic tensor → int64 hash (reference hash.py:75)."""

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        flat = x.reshape(x.shape[0], -1) if x.dim() > 1 else x.reshape(-1, 1)
        out = torch.empty(flat.shape[0], dtype=torch.int64)
        for i in range(flat.shape[0]):
            h = hashlib.sha256(flat[i].cpu().numpy().tobytes()).digest()[:8]
            out[i] = int.from_bytes(h, "little", signed=True)
        return out


class RandomProjectionHash:
    """LSH: random projection + sign bits (reference hash.py:119)."""

    def __init__(self, n_components: int = 16, seed: int = 0):
        self.n_components = n_components
        self.seed = seed
        self._proj: Optional[torch.Tensor] = None

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        flat = x.reshape(x.shape[0], -1).float()
        if self._proj is None or self._proj.shape[0] != flat.shape[1]:
            gen = torch.Generator().manual_seed(self.seed)
            self._proj = torch.randn(flat.shape[1], self.n_components, generator=gen)
        bits = (flat @ self._proj.to(flat.device)) > 0
        weights = (2 ** torch.arange(self.n_components, device=flat.device)).long()
        return (bits.long() * weights).sum(-1)


class QueryModule:
    """Map selected td keys → hashed index (reference query.py:59)."""

    def __init__(self, in_keys: Sequence, index_key: str = "_index", hash_module: Optional[Callable] = None):
        self.in_keys = list(in_keys)
        self.index_key = index_key
        self.hash_module = hash_module or SipHash()

    def __call__(self, td: TensorDictBase) -> TensorDictBase:
        parts = [td.get(k).reshape(td.batch_size[0] if td.batch_size else 1, -1) for k in self.in_keys]
        key = torch.cat(parts, -1)
        td.set(self.index_key, self.hash_module(key))
        return td


class TensorDictMap:
    """Dict keyed by hashed TensorDict entries (reference tdstorage.py:59)."""

    def __init__(self, query_module: QueryModule):
        self.query = query_module
        self._store: Dict[int, TensorDictBase] = {}

    def __setitem__(self, key_td: TensorDictBase, value: TensorDictBase):
        key_td = self.query(key_td.clone(False))
        idx = key_td.get(self.query.index_key)
        for i, h in enumerate(idx.tolist()):
            self._store[h] = value[i] if value.batch_size else value

    def __getitem__(self, key_td: TensorDictBase):
        key_td = self.query(key_td.clone(False))
        idx = key_td.get(self.query.index_key)
        outs = [self._store[h] for h in idx.tolist()]
        return td_stack(outs, 0) if len(outs) > 1 else outs[0]

    def __contains__(self, key_td: TensorDictBase) -> bool:
        key_td = self.query(key_td.clone(False))
        idx = key_td.get(self.query.index_key)
        return all(h in self._store for h in idx.tolist())

    def __len__(self):
        return len(self._store)


class Tree:
    """MCTS node (reference tree.py:30)."""

    __slots__ = ("state", "children", "visits", "value_sum", "prior", "parent", "action")

    def __init__(self, state: Optional[TensorDictBase] = None, prior: float = 0.0, parent=None, action=None):
        self.state = state
        self.children: Dict[Any, "Tree"] = {}
        self.visits = 0
        self.value_sum = 0.0
        self.prior = prior
        self.parent = parent
        self.action = action

    @property
    def value(self) -> float:
        return self.value_sum / self.visits if self.visits else 0.0

    def expand(self, actions, priors=None):
        for i, a in enumerate(actions):
            key = a if not isinstance(a, torch.Tensor) else int(a.item()) if a.numel() == 1 else tuple(a.tolist())
            self.children[key] = Tree(
                prior=float(priors[i]) if priors is not None else 0.0,
                parent=self,
                action=a,
            )

    def backup(self, value: float):
        node = self
        while node is not None:
            node.visits += 1
            node.value_sum += value
            node = node.parent


class PUCTScore:
    """AlphaZero-style selection score (reference scores.py:34)."""

    def __init__(self, c: float = 1.0):
        self.c = c

    def __call__(self, node: Tree) -> Dict[Any, float]:
        total = max(1, node.visits)
        return {
            a: ch.value + self.c * ch.prior * (total**0.5) / (1 + ch.visits)
            for a, ch in node.children.items()
        }


class EXP3Score:
    """EXP3 adversarial-bandit selection (reference mcts/scores.py:241):
    exponential weights with a gamma-mixed uniform exploration floor."""

    def __init__(self, gamma: float = 0.1):
        if not 0 <= gamma <= 1:
            raise ValueError("gamma must be in [0, 1]")
        self.gamma = gamma

    def __call__(self, node: "Tree") -> Dict[Any, float]:
        import math

        actions = list(node.children.keys())
        k = len(actions)
        if k == 0:
            return {}
        # weight = exp(gamma * mean_reward * visits / k), probability-mixed
        ws = {
            a: math.exp(min(50.0, self.gamma * ch.value * max(1, ch.visits) / k))
            for a, ch in node.children.items()
        }
        total = sum(ws.values())
        return {
            a: (1 - self.gamma) * w / total + self.gamma / k
            for a, w in ws.items()
        }


class UCBScore:
    """UCB1 (reference scores.py:150)."""

    def __init__(self, c: float = 1.41):
        self.c = c

    def __call__(self, node: Tree) -> Dict[Any, float]:
        import math

        total = max(1, node.visits)
        out = {}
        for a, ch in node.children.items():
            if ch.visits == 0:
                out[a] = float("inf")
            else:
                out[a] = ch.value + self.c * math.sqrt(math.log(total) / ch.visits)
        return out


class MCTSForest:
    """Collection of search trees keyed by root state
    (reference tree.py:682).

    ``extend(rollout)`` adds observed transitions; ``get_tree(root)``
    returns (building if needed) the tree rooted at a state.
    """

    def __init__(self, score=None):
        self.score = score or UCBScore()
        self._roots: Dict[int, Tree] = {}
        self._hash = SipHash()

    def _key(self, td: TensorDictBase) -> int:
        obs = td.get("observation")
        return int(self._hash(obs.reshape(1, -1))[0])

    def get_tree(self, root_td: TensorDictBase) -> Tree:
        k = self._key(root_td)
        if k not in self._roots:
            self._roots[k] = Tree(state=root_td.clone(False))
        return self._roots[k]

    def extend(self, rollout: TensorDictBase) -> None:
        """Record a [T]-rollout into the forest as a path from its root."""
        if rollout.batch_dims != 1:
            rollout = rollout.reshape(-1)
        root = self.get_tree(rollout[0])
        node = root
        T = rollout.batch_size[0]
        ret = float(rollout.get(("next", "reward")).sum())
        for t in range(T):
            a = rollout[t].get("action")
            key = int(a.argmax().item()) if a.dtype != torch.int64 or a.numel() > 1 else int(a.item())
            if key not in node.children:
                node.children[key] = Tree(parent=node, action=a)
            node = node.children[key]
            node.state = rollout[t].get("next").clone(False)
        node.backup(ret)

    def select(self, node: Tree):
        scores = self.score(node)
        return max(scores, key=scores.get)

    def __len__(self):
        return len(self._roots)


class BinaryToDecimal(torch.nn.Module):
    """Convert binary-encoded tensors to decimals (reference
    map/hash.py:13): groups of ``num_bits`` bits fold to integers,
    summed over the last dim when the input is wider."""

    def __init__(self, num_bits: int, device=None, dtype=torch.int64, convert_to_binary: bool = True):
        super().__init__()
        self.num_bits = num_bits
        self.convert_to_binary = convert_to_binary
        self.register_buffer(
            "bases", 2 ** torch.arange(num_bits - 1, -1, -1, device=device)
        )
        self.dtype = dtype

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.convert_to_binary:
            x = torch.heaviside(x.float(), torch.zeros((), device=x.device))
        L = x.shape[-1]
        if L % self.num_bits:
            raise ValueError("input length must divide num_bits")
        g = x.reshape(*x.shape[:-1], L // self.num_bits, self.num_bits)
        vals = (g.to(self.dtype) * self.bases.to(self.dtype)).sum(-1)
        return vals.sum(-1) if vals.shape[-1] > 1 else vals.squeeze(-1)


class HashToInt:
    """Map arbitrary hash values to compact consecutive ints (reference
    map/hash.py HashToInt) — stable indices for storage-backed maps."""

    def __init__(self):
        self._map: Dict[int, int] = {}

    def __call__(self, key: torch.Tensor) -> torch.Tensor:
        flat = key.reshape(-1)
        out = torch.empty_like(flat)
        for i, v in enumerate(flat.tolist()):
            out[i] = self._map.setdefault(int(v), len(self._map))
        return out.reshape(key.shape)

    def state_dict(self):
        return {"map": dict(self._map)}

    def load_state_dict(self, sd):
        self._map = {int(k): int(v) for k, v in sd["map"].items()}
