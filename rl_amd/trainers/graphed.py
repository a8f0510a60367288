"""GraphedPPO — whole-iteration hipGraph PPO training step.

The flagship MI355X training loop, built ONLY from public library
components: a :class:`~rl_amd.collectors.Collector` (whose GPU fast path
runs the rollout as one hipGraph replay or a single mega-kernel launch),
a :class:`~rl_amd.objectives.value.advantages.GAE` estimator on the fused
HIP scan, a :class:`~rl_amd.objectives.ClipPPOLoss`, and an optimizer.

Single GPU: the ENTIRE iteration — rollout + GAE + every minibatch
forward/backward + the (capturable) optimizer — is captured once as ONE
hipGraph and replayed per step, removing the launch/latency floor
(measured 26M frames/s at 4096 envs × T=16, profiles/README.md).

Distributed (one process per GPU over RCCL/xGMI): the whole step cannot
be one graph (the gradient all-reduce sits between backward and the
optimizer), so the launch-heavy minibatch forward+backward is captured as
its own graph and the bucketed :class:`~rl_amd.parallel.comm.GradAllReducer`
runs on a second HIP stream — async all-reduce per bucket, applied just
before the optimizer step.  When the minibatch capture is unavailable the
reducer's post-accumulate-grad hooks overlap the all-reduce with the
eager backward instead.

Reference analog: pytorch/rl trainers/trainers.py:1400 (train loop) +
trainers/_distributed.py:138 (DDP learner); rebuilt MI355X-first around
graph capture instead of a module wrapper.
"""
from __future__ import annotations

import sys
from typing import Callable, Optional

import torch

from ..tensordict import TensorDictBase

__all__ = ["GraphedPPO"]


class GraphedPPO:
    """One-call-per-iteration PPO training step with hipGraph capture.

    Args:
        collector: a :class:`rl_amd.collectors.Collector` whose
            ``frames_per_batch`` equals ``n_envs * horizon``.
        advantage: a value estimator module (e.g. ``GAE``) called on the
            rollout batch under ``no_grad``.
        loss_module: loss whose forward returns ``loss_*`` keys summed
            into the total objective (e.g. ``ClipPPOLoss``).
        optimizer: optimizer over the trained parameters.  For full-step
            capture it must be capturable (``torch.optim.Adam(...,
            capturable=True, fused=True)``).
        minibatches / epochs: PPO update schedule per iteration.
        autocast_dtype: compute dtype for loss forward/backward.
        max_grad_norm: gradient clipping threshold (0 disables).
        capture: ``"auto"`` (full-step graph when single-process CUDA and
            the store fits), ``True`` (require), ``False`` (eager).
        reducer: optional :class:`GradAllReducer`; built automatically
            when ``torch.distributed`` is initialized with world > 1.
        post_optim_hook: called after every optimizer step (e.g.
            ``refresh_splitk_caches``).
        max_capture_frames: full-step capture is disabled above this
            store size (the capture memory pool cored at 65536×64 frames
            on MI355X; the mega-kernel rollout carries those sizes).
    """

    def __init__(
        self,
        collector,
        advantage,
        loss_module,
        optimizer,
        *,
        minibatches: int = 4,
        epochs: int = 1,
        autocast_dtype: torch.dtype = torch.bfloat16,
        max_grad_norm: float = 1.0,
        capture="auto",
        reducer=None,
        post_optim_hook: Optional[Callable[[], None]] = None,
        max_capture_frames: int = 2_000_000,
    ):
        self.collector = collector
        self.advantage = advantage
        self.loss_module = loss_module
        self.optimizer = optimizer
        self.minibatches = minibatches
        self.epochs = epochs
        self.autocast_dtype = autocast_dtype
        self.max_grad_norm = max_grad_norm
        self.capture = capture
        self.post_optim_hook = post_optim_hook
        self.max_capture_frames = max_capture_frames

        self._cuda = torch.cuda.is_available()
        self._device = None
        self._params = [
            p
            for group in optimizer.param_groups
            for p in group["params"]
            if p.requires_grad
        ]
        import torch.distributed as dist

        self._world = dist.get_world_size() if dist.is_initialized() else 1
        self.reducer = reducer
        if self.reducer is None and self._world > 1:
            from ..parallel.comm import GradAllReducer

            self.reducer = GradAllReducer(self._params, world_size=self._world)
        self._distributed = self.reducer is not None and self._world > 1
        # test hook: exercise the distributed-mode step shape (minibatch
        # graph + eager comm/step) at world=1 — used to argue multi-GPU
        # per-rank throughput from single-GPU measurements
        import os

        if os.environ.get("RL_AMD_FORCE_MB_GRAPH") == "1":
            self._distributed = True

        self._seed_one = None
        self._clip_scale = None
        self._gradsq_part = None
        if (
            self._cuda
            and max_grad_norm
            and all(g.get("fused") for g in optimizer.param_groups)
            and isinstance(optimizer, torch.optim.Adam)
        ):
            try:
                dev = next(p.device for p in self._params)
                self._clip_scale = torch.ones((), device=dev)
                optimizer.grad_scale = self._clip_scale
            except Exception:
                self._clip_scale = None
        self._initialized = False
        self._step_fn: Optional[Callable[[], None]] = None
        self._full_graph = False
        self._mb_graph = None
        self._mb_static: Optional[TensorDictBase] = None
        self._autocast = torch.autocast(
            device_type="cuda",
            dtype=autocast_dtype,
            enabled=self._cuda,
            cache_enabled=False,  # the autocast weight cache allocates
            # mid-capture, which hipGraph capture forbids
        )

    # ------------------------------------------------------------------ #
    def _total_loss(self, out: TensorDictBase) -> torch.Tensor:
        # kernel-side pre-summed losses from the ClipPPOLoss mega path:
        # "_loss_total" = objective + entropy + scaled critic;
        # "_loss_actor" = objective + entropy
        if hasattr(out, "get"):
            full = out.get("_loss_total", None)
            if full is not None:
                skip = {"loss_objective", "loss_entropy", "loss_critic"}
                total = full
                for k in out.keys():
                    if (isinstance(k, str) and k.startswith("loss_")
                            and k not in skip):
                        total = total + out.get(k)
                return total
            presummed = out.get("_loss_actor", None)
        else:
            presummed = None
        skip = {"loss_objective", "loss_entropy"} if presummed is not None else set()
        total = presummed
        for k in out.keys():
            if isinstance(k, str) and k.startswith("loss_") and k not in skip:
                v = out.get(k)
                total = v if total is None else total + v
        return total

    def _mb_fwd_bwd(self, sub: TensorDictBase) -> None:
        with self._autocast:
            out = self.loss_module(sub)
            total = self._total_loss(out)
        # set_to_none=True even under capture: grads allocated inside the
        # capture live in the graph's private pool (stable across
        # replays), and it removes a zero-fill + accumulate-add per
        # parameter per minibatch (~100 kernels/step in the T=16 profile)
        self.optimizer.zero_grad(set_to_none=True)
        if total.is_cuda:
            if self._seed_one is None or self._seed_one.device != total.device:
                self._seed_one = torch.ones((), device=total.device)
            # preallocated seed: backward() otherwise fills a fresh ones
            # scalar every minibatch
            total.backward(gradient=self._seed_one)
            # join any loss-module side streams (e.g. the ClipPPOLoss
            # critic branch) before gradients are read
            for s in getattr(self.loss_module, "_side_streams", []):
                torch.cuda.current_stream().wait_stream(s)
        else:
            total.backward()

    def _capture_mb_graph(self, example_sub: TensorDictBase) -> None:
        static_sub = example_sub.clone(False)
        for k in list(static_sub.keys(True, True)):
            static_sub.set(k, static_sub.get(k).clone())
        if self.reducer is not None:
            self.reducer.hooks_enabled = False  # a hook firing during
            # capture would record the collective into the graph
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                self._mb_fwd_bwd(static_sub)
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._mb_fwd_bwd(static_sub)
        self._mb_graph = g
        self._mb_static = static_sub

    def _run_minibatch(self, sub: TensorDictBase) -> None:
        use_mb_graph = self._cuda and self._distributed and self.capture is not False
        if use_mb_graph and self._mb_graph is None and self._mb_graph is not False:
            try:
                self._capture_mb_graph(sub)
            except Exception:
                import traceback

                traceback.print_exc(file=sys.stderr)
                self._mb_graph = False
                if self.reducer is not None:
                    self.reducer.hooks_enabled = True
        if use_mb_graph and self._mb_graph not in (None, False):
            static_sub = self._mb_static
            for k in list(static_sub.keys(True, True)):
                static_sub.get(k).copy_(sub.get(k))
            self._mb_graph.replay()
            if self.reducer is not None:
                # backward was a graph replay: hooks cannot fire, launch
                # every bucket explicitly (still async, comm stream)
                self.reducer.reduce()
        else:
            self._mb_fwd_bwd(sub)
            # hook-mode reducer already launched its buckets during
            # backward (overlapped); nothing to do here
        if self.reducer is not None:
            self.reducer.finalize()
        if self.max_grad_norm:
            from .. import ops

            if (
                self._clip_scale is not None
                and self.loss_module.__dict__.pop("_gradsq_armed", False)
            ):
                # the merged-loss backward already left grad-sumsq
                # partials in _gradsq_part: one tiny finalize kernel
                from .. import _C

                _C.wgrad_clip_finalize(
                    self._gradsq_part, self.max_grad_norm,
                    self._clip_scale, True,
                )
            elif self._clip_scale is not None:
                # the clip coefficient rides into the fused Adam as its
                # grad_scale divisor: no gradient multiply at all
                ops.fused_grad_clip_scale_(
                    self._params, self.max_grad_norm, self._clip_scale
                )
            elif not ops.fused_grad_clip_(self._params, self.max_grad_norm):
                torch.nn.utils.clip_grad_norm_(self._params, self.max_grad_norm)
        self.optimizer.step()
        if self.post_optim_hook is not None:
            self.post_optim_hook()

    def _minibatch_keys(self, flat: TensorDictBase):
        """Keys the update phase actually reads (loss tensor_keys +
        network in_keys); shuffling anything else is wasted gather
        bandwidth.  None = shuffle everything (unknown loss shape)."""
        if "_mb_keys" in self.__dict__:
            return self.__dict__["_mb_keys"]
        keys = None
        try:
            tk = self.loss_module.tensor_keys
            keys = set()
            attrs = ["advantage", "value_target", "action", "sample_log_prob"]
            if getattr(self.loss_module, "clip_value", None) is not None:
                attrs.append("value")  # old values only needed for clipping
            for attr in attrs:
                k = getattr(tk, attr, None)
                if k is not None:
                    keys.add(k)
            for net_attr in ("actor_network", "critic_network"):
                net = getattr(self.loss_module, net_attr, None)
                if net is not None:
                    keys.update(net.in_keys)
            present = set(flat.keys(include_nested=True, leaves_only=True))
            keys = [k for k in keys if k in present]
            needed_min = {"advantage", "action"}
            if not needed_min.issubset({k if isinstance(k, str) else k[-1]
                                        for k in keys}):
                keys = None
        except Exception:
            keys = None
        self.__dict__["_mb_keys"] = keys
        return keys

    def _update_phase(self, batch: TensorDictBase) -> None:
        with torch.no_grad(), self._autocast:
            self.advantage(batch)
        flat = batch.reshape(-1)
        keys = self._minibatch_keys(flat)
        if keys is not None:
            flat = flat.select(*keys)
        n = flat.batch_size[0]
        mb = n // self.minibatches
        device = flat.device
        from .. import ops

        on_gpu = any(
            isinstance(v, torch.Tensor) and v.is_cuda for v in flat.values()
        )
        for _ in range(self.epochs):
            # one shuffle-gather of the whole flat store, then
            # minibatches are contiguous zero-copy slices.  On GPU the
            # shuffle is ONE kernel applying a keyed Feistel
            # permutation inline (fresh philox keys per epoch);
            # randperm's radix sort + per-key index kernels otherwise.
            shuffled = None
            if on_gpu:
                keys = torch.randint(
                    -(2 ** 31), 2 ** 31 - 1, (4,), device=device,
                    dtype=torch.int32,
                )
                shuffled = ops.multi_shuffle_td(flat, keys)
            if shuffled is None:
                perm = torch.randperm(n, device=device)
                shuffled = ops.multi_gather_td(flat, perm) if on_gpu else None
                if shuffled is None:
                    shuffled = flat[perm]
            # one batched launch computes every minibatch's advantage
            # normalization stats (the loss would otherwise launch a
            # stats pair per minibatch).  Not in mb-graph mode: the
            # capture would bake the first epoch's stats addresses.
            stats_all = None
            eps_all = None
            if (
                on_gpu
                and not self._distributed
                and getattr(self.loss_module, "normalize_advantage", False)
                and ops.HAS_HIP_EXT
                and n % self.minibatches == 0
            ):
                try:
                    advk = self.loss_module.tensor_keys.advantage
                    adv_t = shuffled.get(advk, None)
                    if adv_t is not None and adv_t.dtype == torch.float32:
                        stats_all = ops.adv_stats_batch(
                            adv_t.reshape(-1), self.minibatches
                        )
                    if (
                        self._clip_scale is not None
                        and self.reducer is None
                        and self._gradsq_part is None
                        # the sq partials only cover the merged-loss
                        # Function's 12 gradients (2 nets x 3 linears)
                        and len(self._params) == 12
                    ):
                        self._gradsq_part = torch.empty(
                            6 * 512, device=device
                        )
                    if self._gradsq_part is not None:
                        self.loss_module.__dict__["_gradsq_part"] = (
                            self._gradsq_part
                        )
                    # one philox draw for every minibatch's entropy eps
                    ak = self.loss_module.tensor_keys.action
                    act_t = shuffled.get(ak, None)
                    if act_t is not None:
                        eps_all = torch.randn(
                            self.minibatches, mb, act_t.shape[-1],
                            device=device, dtype=torch.float32,
                        )
                except Exception:
                    stats_all = eps_all = None
            for i in range(self.minibatches):
                if stats_all is not None:
                    self.loss_module.__dict__["_mega_stats"] = stats_all[i]
                if eps_all is not None:
                    self.loss_module.__dict__["_mega_eps"] = eps_all[i]
                self._run_minibatch(shuffled[i * mb : (i + 1) * mb])
            if stats_all is not None:
                self.loss_module.__dict__.pop("_mega_stats", None)
            if eps_all is not None:
                self.loss_module.__dict__.pop("_mega_eps", None)

    def _one_iter_inline(self) -> None:
        batch = self.collector.rollout_inline()
        self._update_phase(batch)

    def _one_iter(self) -> None:
        batch = self.collector.rollout()
        self._update_phase(batch)

    # ------------------------------------------------------------------ #
    def initialize(self) -> "GraphedPPO":
        """Warm up and (when eligible) capture the full-step graph."""
        if self._initialized:
            return self
        self._initialized = True
        want_full = (
            self.capture is not False
            and self._cuda
            and not self._distributed
            and self.collector.frames_per_batch <= self.max_capture_frames
        )
        if want_full:
            try:
                # materialize the collector fast path, the rollout store
                # and the GAE/loss buffers before any stream juggling
                self._one_iter_inline()
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(3):
                        self._one_iter_inline()
                torch.cuda.current_stream().wait_stream(side)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._one_iter_inline()
                self._step_fn = g.replay
                self._full_graph = True
            except Exception as e:
                import traceback

                traceback.print_exc(file=sys.stderr)
                print(
                    f"[GraphedPPO] full-step capture failed ({e!r}); "
                    "falling back to per-phase execution",
                    file=sys.stderr,
                )
                if self.capture is True:
                    raise
                self._step_fn = self._one_iter
        else:
            self._step_fn = self._one_iter
        return self

    @property
    def full_graph(self) -> bool:
        return self._full_graph

    @property
    def minibatch_graph(self) -> bool:
        return self._mb_graph not in (None, False)

    def step(self) -> None:
        """Run one full PPO iteration (rollout → GAE → updates)."""
        if not self._initialized:
            self.initialize()
        self._step_fn()
