"""EnvCreator — picklable env factory with shared-state propagation.

Reference: pytorch/rl torchrl/envs/env_creator.py:20: wraps an env
constructor so worker processes can rebuild the env AND share stateful
transform buffers (VecNorm running stats) with the parent.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Optional

from ..tensordict import TensorDictBase

__all__ = ["EnvCreator", "env_creator"]


class EnvCreator:
    def __init__(self, create_env_fn: Callable, create_env_kwargs: Optional[dict] = None):
        self.create_env_fn = create_env_fn
        self.create_env_kwargs = create_env_kwargs or {}
        self._shared_td: Optional[TensorDictBase] = None
        self.init_()

    def init_(self) -> None:
        """Build one env to harvest shareable state (VecNorm stats etc.),
        put it in shared memory, and keep it for propagation."""
        env = self.create_env_fn(**self.create_env_kwargs)
        shared = self._harvest_shared(env)
        if shared is not None:
            shared.share_memory_()
        self._shared_td = shared
        if hasattr(env, "close"):
            env.close()

    @staticmethod
    def _harvest_shared(env) -> Optional[TensorDictBase]:
        transform = getattr(env, "transform", None)
        if transform is None:
            return None
        from .transforms.vecnorm import VecNorm

        for t in getattr(transform, "transforms", [transform]):
            if isinstance(t, VecNorm) and not t._stats.is_empty():
                return t._stats
        return None

    def __call__(self):
        env = self.create_env_fn(**self.create_env_kwargs)
        if self._shared_td is not None:
            from .transforms.vecnorm import VecNorm

            transform = getattr(env, "transform", None)
            if transform is not None:
                for t in getattr(transform, "transforms", [transform]):
                    if isinstance(t, VecNorm):
                        t._stats = self._shared_td
        return env

    def __repr__(self):
        return f"EnvCreator({self.create_env_fn})"


def env_creator(fn: Callable) -> EnvCreator:
    return EnvCreator(fn)


def get_env_metadata(env_or_creator, kwargs=None):
    """Extract picklable metadata (specs, batch size, device) from an
    env or an env-making callable (reference env_creator.py): batched /
    remote workers rebuild specs from this without instantiating the
    env again."""
    from .common import EnvMetaData

    from .common import EnvBase

    if isinstance(env_or_creator, EnvBase):
        env = env_or_creator
    else:
        env = env_or_creator(**(kwargs or {}))
    return EnvMetaData.build(env)


__all__.append("get_env_metadata")
