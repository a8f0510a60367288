"""Collector (= SyncDataCollector) — the single-process execution engine.

Reference: pytorch/rl torchrl/collectors/_single.py:297 (Collector; hot
loop ``rollout``:2014, iterator:1761, carrier handling:1388), naming per
torchrl/collectors/__init__.py:27-47 (``Collector`` IS the old
``SyncDataCollector``).

MI355X design: with a GPU-resident vectorized env and a GPU policy the
whole rollout loop stays on-device — the carrier TensorDict never crosses
PCIe, policy calls and env steps queue on one HIP stream, and the output
batch is assembled by indexed writes into a pre-allocated [B, T] buffer in
HBM.
"""
from __future__ import annotations

import time
from typing import Any, Callable, Iterator, Optional, Sequence, Union

import torch

from .._utils import logger, prod, timeit
from ..envs.common import EnvBase
from ..envs.utils import ExplorationType, RandomPolicy, set_exploration_type, step_mdp
from ..tensordict import TensorDict, TensorDictBase, stack as td_stack
from .utils import split_trajectories

__all__ = ["Collector", "SyncDataCollector", "BaseCollector"]


class BaseCollector:
    """ABC for collectors (reference torchrl/collectors/_base.py:220)."""

    def __iter__(self) -> Iterator[TensorDictBase]:
        return self.iterator()

    def iterator(self):
        raise NotImplementedError

    def register_weight_sync_scheme(self, model_id: str, scheme, model=None) -> None:
        """Register a WeightSyncScheme for ``model_id`` (reference
        per-model_id scheme registry, weight_sync_schemes.py:346); its
        sender is driven by ``update_policy_weights_(model_id=...)``."""
        if not hasattr(self, "_weight_senders"):
            self._weight_senders = {}
        target = model if model is not None else getattr(self, "policy", None)
        self._weight_senders[model_id] = scheme.create_sender(model_id, target)

    def update_policy_weights_(self, policy_or_weights=None, *, model_id=None, **kwargs) -> None:
        """Push new policy weights into the collector (reference
        _base.py:924).  With registered schemes, route through the
        model_id's sender; single-process default copies the state-dict
        into the local policy."""
        senders = getattr(self, "_weight_senders", None)
        if senders:
            if model_id is not None:
                senders[model_id].send(policy_or_weights)
            else:
                for sender in senders.values():
                    sender.send(policy_or_weights)
            return
        if policy_or_weights is None:
            return
        if isinstance(policy_or_weights, dict):
            self.policy.load_state_dict(policy_or_weights)
        elif isinstance(policy_or_weights, TensorDictBase):
            policy_or_weights.to_module(self.policy)
        elif hasattr(policy_or_weights, "state_dict"):
            self.policy.load_state_dict(policy_or_weights.state_dict())

    def set_seed(self, seed: int, static_seed: bool = False) -> int:
        raise NotImplementedError

    def state_dict(self) -> dict:
        return {}

    def load_state_dict(self, sd: dict) -> None:
        pass

    def shutdown(self, timeout: Optional[float] = None) -> None:
        pass

    def stats(self) -> dict:
        return {}

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.shutdown()


class Collector(BaseCollector):
    """Single-process synchronous collector.

    Iterating yields TensorDict batches of ``frames_per_batch`` frames
    shaped ``[B, T]`` (batched env) or ``[T]``.
    """

    def __init__(
        self,
        create_env_fn: Union[EnvBase, Callable[[], EnvBase]],
        policy: Optional[Callable[[TensorDictBase], TensorDictBase]] = None,
        *,
        frames_per_batch: int,
        total_frames: int = -1,
        init_random_frames: int = 0,
        device=None,
        storing_device=None,
        policy_device=None,
        env_device=None,
        reset_at_each_iter: bool = False,
        postproc: Optional[Callable] = None,
        split_trajs: bool = False,
        exploration_type: ExplorationType = ExplorationType.RANDOM,
        return_same_td: bool = False,
        reset_when_done: bool = True,
        interruptor=None,
        set_truncated: bool = False,
        replay_buffer=None,
        extend_buffer: bool = True,
        trust_policy: bool = True,
        compile_policy: bool = False,
        cudagraph_policy: bool = False,
        no_cuda_sync: bool = False,
        weight_updater=None,
        use_graph: Union[bool, str] = "auto",
        **kwargs,
    ):
        if isinstance(create_env_fn, EnvBase):
            self.env = create_env_fn
        else:
            self.env = create_env_fn()
        self.closed = False
        device = torch.device(device) if device is not None else None
        self.storing_device = (
            torch.device(storing_device) if storing_device is not None else device
        )
        self.policy_device = (
            torch.device(policy_device) if policy_device is not None else device
        )
        self.env_device = (
            torch.device(env_device) if env_device is not None else (device or self.env.device)
        )
        if device is not None and self.env.device != self.env_device:
            self.env = self.env.to(self.env_device)
        if policy is None:
            policy = RandomPolicy(self.env.full_action_spec)
        self.policy = policy
        if self.policy_device is not None and hasattr(policy, "to"):
            self.policy = policy.to(self.policy_device)
        if compile_policy and hasattr(self.policy, "forward"):
            self.policy = torch.compile(self.policy)
        self._cast_policy = (
            self.policy_device is not None
            and self.env_device is not None
            and self.policy_device != self.env_device
        )

        self.env_batch = self.env.batch_size
        n_envs = max(1, prod(self.env_batch))
        if frames_per_batch % n_envs != 0:
            raise ValueError(
                f"frames_per_batch ({frames_per_batch}) must divide evenly by the "
                f"number of envs ({n_envs})"
            )
        self.n_envs = n_envs
        self.frames_per_batch = frames_per_batch
        self.steps_per_batch = frames_per_batch // n_envs
        self.total_frames = total_frames if total_frames > 0 else float("inf")
        self.init_random_frames = init_random_frames
        self.reset_at_each_iter = reset_at_each_iter
        self.postproc = postproc
        self.split_trajs = split_trajs
        self.exploration_type = exploration_type
        self.return_same_td = return_same_td
        self.replay_buffer = replay_buffer
        self.extend_buffer = extend_buffer
        self.set_truncated = set_truncated
        self.interruptor = interruptor

        self._frames = 0
        self._iter = 0
        # GPU fast path: hipGraph-captured (or mega-kernel) rollout into a
        # static [B, T] HBM store — engaged lazily on the first rollout
        self.use_graph = use_graph
        self._graphed = None
        self._graphed_tried = False
        # persistent carrier (reference _make_carrier:1388)
        self._carrier: Optional[TensorDictBase] = None
        # trajectory ids (reference _update_traj_ids:1985)
        self._traj_pool_top = n_envs
        self._traj_ids = torch.arange(n_envs).reshape(self.env_batch or (1,))
        if not self.env_batch:
            self._traj_ids = self._traj_ids.reshape(())
        self._stats = {"frames": 0, "iter_time": 0.0}

    # ------------------------------------------------------------------ #
    @property
    def frames(self) -> int:
        return self._frames

    def set_seed(self, seed: int, static_seed: bool = False) -> int:
        return self.env.set_seed(seed, static_seed=static_seed)

    def _update_traj_ids(self, done: torch.Tensor) -> None:
        """Assign fresh trajectory ids to envs that finished."""
        done_flat = done.reshape(self._traj_ids.shape)
        n_done = int(done_flat.sum())
        if n_done:
            new_ids = torch.arange(
                self._traj_pool_top, self._traj_pool_top + n_done
            )
            self._traj_pool_top += n_done
            ids = self._traj_ids.reshape(-1).clone()
            ids[done_flat.reshape(-1)] = new_ids
            self._traj_ids = ids.reshape(self._traj_ids.shape)

    def _try_graphed(self):
        """Build the GraphedRollout fast path when eligible: GPU-resident
        vectorized env with masked auto-reset + GPU policy.  The whole
        T-step loop then runs as one hipGraph replay (or, when the env
        advertises ``supports_fused_rollout`` for the policy, as a single
        mega-kernel launch) writing straight into a static [B, T] HBM
        store.  The returned batch is that store, overwritten per
        iteration (``return_same_td`` semantics); ('collector','traj_ids')
        is not stamped on this path."""
        if not self.use_graph:
            return None
        if not (
            torch.cuda.is_available()
            and self.env.device is not None
            and self.env.device.type == "cuda"
            and len(self.env_batch) == 1
            and self.steps_per_batch >= 1
            and self.interruptor is None
            and self.init_random_frames == 0
            and not self.reset_at_each_iter
            and (self.storing_device is None or self.storing_device == self.env.device)
            and not self._cast_policy
            and getattr(self.env, "_supports_masked_reset", False)
        ):
            return None
        policy = self.policy
        try:
            from ..ops import HAS_HIP_EXT, FusedTanhNormalActor

            if HAS_HIP_EXT and not isinstance(policy, FusedTanhNormalActor):
                try:
                    policy = FusedTanhNormalActor(policy)
                except Exception:
                    policy = self.policy
        except Exception:
            pass
        try:
            from .graph import GraphedRollout

            gr = GraphedRollout(
                self.env,
                policy,
                horizon=self.steps_per_batch,
                exploration_type=self.exploration_type,
            ).initialize()
        except Exception:
            if self.use_graph is True:
                raise
            return None
        if not (gr.mega or gr.captured) and self.use_graph == "auto":
            # neither capture nor the mega-kernel engaged: the graphed
            # body would just be the eager loop without collector
            # metadata — keep the standard path
            return None
        return gr

    def rollout_inline(self) -> TensorDictBase:
        """``rollout`` variant safe to run inside an OUTER hipGraph
        capture: the fast path runs its uncaptured body (a graph cannot
        replay another graph).  Used by trainers.GraphedPPO."""
        if not self._graphed_tried:
            self._graphed_tried = True
            self._graphed = self._try_graphed()
        if self._graphed is not None:
            self._graphed.body()
            self._frames += self.frames_per_batch
            return self._graphed.store
        return self.rollout()

    def rollout(self) -> TensorDictBase:
        """The per-step hot loop (reference _single.py:2014)."""
        if not self._graphed_tried:
            self._graphed_tried = True
            self._graphed = self._try_graphed()
        if self._graphed is not None:
            batch = self._graphed.collect()
            self._frames += self.frames_per_batch
            return batch
        if self._carrier is None or self.reset_at_each_iter:
            self._carrier = self.env.reset()
        carrier = self._carrier
        snaps = []
        t_dim = len(self.env_batch)
        with set_exploration_type(self.exploration_type), torch.no_grad():
            for t in range(self.steps_per_batch):
                if self._frames < self.init_random_frames:
                    carrier = self.env.rand_action(carrier)
                else:
                    if self._cast_policy:
                        carrier = carrier.to(self.policy_device)
                    carrier = self.policy(carrier)
                    if self._cast_policy:
                        carrier = carrier.to(self.env_device)
                carrier, next_root = self.env.step_and_maybe_reset(carrier)
                # stamp collector metadata
                carrier.set(
                    ("collector", "traj_ids"),
                    self._traj_ids.to(carrier.device) if carrier.device else self._traj_ids,
                )
                snap = carrier.clone(False)
                if self.storing_device is not None and snap.device != self.storing_device:
                    snap = snap.to(self.storing_device)
                snaps.append(snap)
                done = carrier.get(("next", "done"))
                self._update_traj_ids(done)
                carrier = next_root
                self._frames += self.n_envs
                if self.interruptor is not None and self.interruptor.collection_stopped():
                    break
        self._carrier = carrier
        batch = td_stack(snaps, t_dim)
        return batch

    def iterator(self) -> Iterator[TensorDictBase]:
        while self._frames < self.total_frames:
            t0 = time.perf_counter()
            with timeit("collector/rollout"):
                batch = self.rollout()
            if self.postproc is not None:
                batch = self.postproc(batch)
            if self.split_trajs:
                batch = split_trajectories(batch)
            self._iter += 1
            self._stats["frames"] = self._frames
            self._stats["iter_time"] = time.perf_counter() - t0
            if self.replay_buffer is not None:
                if self.extend_buffer:
                    self.replay_buffer.extend(batch.reshape(-1))
                else:
                    self.replay_buffer.add(batch)
                yield None
            else:
                yield batch

    def start(self):
        """Run collection in a background thread writing into the replay
        buffer (reference _single.py:1854)."""
        import threading

        if self.replay_buffer is None:
            raise RuntimeError("start() requires a replay_buffer")
        self._stop_event = threading.Event()

        def _run():
            for _ in self.iterator():
                if self._stop_event.is_set():
                    break

        self._thread = threading.Thread(target=_run, daemon=True)
        self._thread.start()
        return self._thread

    def async_shutdown(self, timeout: float = 10.0):
        if hasattr(self, "_stop_event"):
            self._stop_event.set()
            self._thread.join(timeout)

    def shutdown(self, timeout: Optional[float] = None) -> None:
        if not self.closed:
            self.async_shutdown() if hasattr(self, "_stop_event") else None
            self.env.close()
            self.closed = True

    def state_dict(self) -> dict:
        sd = {"frames": self._frames, "iter": self._iter}
        if hasattr(self.policy, "state_dict"):
            sd["policy_state_dict"] = self.policy.state_dict()
        if hasattr(self.env, "state_dict"):
            sd["env_state_dict"] = self.env.state_dict()
        return sd

    def load_state_dict(self, sd: dict) -> None:
        self._frames = sd.get("frames", 0)
        self._iter = sd.get("iter", 0)
        if "policy_state_dict" in sd and hasattr(self.policy, "load_state_dict"):
            self.policy.load_state_dict(sd["policy_state_dict"])
        if "env_state_dict" in sd and hasattr(self.env, "load_state_dict"):
            try:
                self.env.load_state_dict(sd["env_state_dict"])
            except Exception:
                pass

    def stats(self) -> dict:
        fps = (
            self.frames_per_batch / self._stats["iter_time"]
            if self._stats["iter_time"]
            else 0.0
        )
        return {**self._stats, "fps": fps}

    def __repr__(self):
        return (
            f"Collector(env={type(self.env).__name__}, "
            f"frames_per_batch={self.frames_per_batch}, frames={self._frames})"
        )


# Reference compat alias (old name)
SyncDataCollector = Collector
