"""SerialEnv / ParallelEnv — batched env execution.

Reference: pytorch/rl torchrl/envs/batched_envs.py (BatchedEnvBase:322,
SerialEnv:1546, ParallelEnv:1918, worker loop
_run_worker_pipe_shared_mem:3322, buffered step path :2405).

MI355X note: GPU-vectorized envs (rl_amd/envs/custom) are the PRIMARY
path — one process, one device, tensor-batched dynamics.  ParallelEnv
exists for CPU simulators: N worker processes step their sub-envs straight
into a shared-memory TensorDict slice (zero-copy to the parent), commands
go over pipes, completion over pipe acks.  When the parent's device is a
GPU the shared buffer is page-locked so the follow-up H2D copy is async.
"""
from __future__ import annotations

import multiprocessing as mp
import os
import time
from typing import Any, Callable, List, Optional, Sequence, Union

import torch

from .._utils import _check_for_faulty_process, _ProcessNoWarn, logger
from ..data.tensor_specs import Composite
from ..tensordict import TensorDict, TensorDictBase, stack as td_stack
from .common import EnvBase, EnvMetaData
from .utils import step_mdp, terminated_or_truncated

__all__ = ["SerialEnv", "ParallelEnv", "BatchedEnvBase"]

BATCHED_PIPE_TIMEOUT = float(os.environ.get("BATCHED_PIPE_TIMEOUT", "60.0"))


class BatchedEnvBase(EnvBase):
    """Common spec plumbing for Serial/Parallel envs."""

    def __init__(
        self,
        num_workers: int,
        create_env_fn: Union[Callable[[], EnvBase], Sequence[Callable[[], EnvBase]]],
        *,
        create_env_kwargs: Optional[Union[dict, Sequence[dict]]] = None,
        device=None,
        shared_memory: bool = True,
        **kwargs,
    ):
        if callable(create_env_fn):
            create_env_fn = [create_env_fn] * num_workers
        if len(create_env_fn) != num_workers:
            raise ValueError("need one env constructor per worker")
        if create_env_kwargs is None:
            create_env_kwargs = [{}] * num_workers
        elif isinstance(create_env_kwargs, dict):
            create_env_kwargs = [create_env_kwargs] * num_workers
        self.create_env_fn = list(create_env_fn)
        self.create_env_kwargs = list(create_env_kwargs)
        self.num_workers = num_workers
        self._dummy_env: Optional[EnvBase] = None
        super().__init__(device=device, batch_size=torch.Size([num_workers]))
        self.shared_memory = shared_memory
        self.is_closed = True

    def _set_specs_from(self, env: EnvBase):
        inner_bs = env.batch_size
        self._inner_batch = inner_bs
        n = self.num_workers

        def ex(spec):
            return spec.expand(n, *spec.shape)

        self.full_observation_spec = ex(env.full_observation_spec.clone())
        self.full_action_spec = ex(env.full_action_spec.clone())
        self.full_reward_spec = ex(env.full_reward_spec.clone())
        self.full_done_spec = ex(env.full_done_spec.clone())


class SerialEnv(BatchedEnvBase):
    """N sub-envs stepped sequentially in-process (reference :1546)."""

    def __init__(self, num_workers: int, create_env_fn, **kwargs):
        super().__init__(num_workers, create_env_fn, **kwargs)
        self._envs: List[EnvBase] = [
            fn(**kw) for fn, kw in zip(self.create_env_fn, self.create_env_kwargs)
        ]
        self._set_specs_from(self._envs[0])
        self.is_closed = False

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        outs = []
        for i, env in enumerate(self._envs):
            sub = None
            if tensordict is not None:
                sub = tensordict[i].clone(False)
                if "_reset" in sub and not bool(sub.get("_reset").any()):
                    # not flagged: keep current state, echo the root obs
                    outs.append(env.maybe_reset(sub))
                    continue
            outs.append(env.reset(sub))
        return td_stack(outs, 0)

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        outs = []
        for i, env in enumerate(self._envs):
            td_i = tensordict[i].clone(False)
            td_i = env.step(td_i)
            outs.append(td_i.get("next"))
        return td_stack(outs, 0)

    def _set_seed(self, seed: Optional[int]):
        out = seed
        for i, env in enumerate(self._envs):
            out = env.set_seed(seed + i if seed is not None else None)
        return out

    def close(self, raise_if_closed: bool = False):
        for env in self._envs:
            env.close()
        self.is_closed = True

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            envs = self.__dict__.get("_envs")
            if envs:
                return getattr(envs[0], name)
            raise


def _copy_into(dst: TensorDictBase, src: TensorDictBase) -> None:
    """Copy matching leaves of src into dst in place (dst is a shared-
    memory slice: only pre-existing keys stay shared)."""
    for k in dst.keys(True, True):
        v = src.get(k, None)
        if v is not None and isinstance(v, torch.Tensor):
            dst.get(k).copy_(v)


def _parallel_worker(
    idx: int,
    pipe,
    env_fn,
    env_kwargs: dict,
    shared_td: Optional[TensorDictBase],
):
    """Worker loop (reference _run_worker_pipe_shared_mem:3322).

    Buffered commands complete via a shared-memory done-flag write
    (reference _signal_done :3390-3397) — the parent spin-waits one
    cache line per worker instead of draining N pipe acks; pipes carry
    only the command tuples and the unbuffered fallback payloads."""
    torch.set_num_threads(1)
    env = env_fn(**env_kwargs)
    my_slice = shared_td[idx] if shared_td is not None else None
    flags: Optional[torch.Tensor] = None
    seq = 0
    pipe.send(("meta", EnvMetaData.build(env)))
    root_td: Optional[TensorDictBase] = None
    action_keys = env.full_action_spec.keys(True, True) or ["action"]

    def _signal_done():
        nonlocal seq
        seq += 1
        flags[idx] = seq

    try:
        while True:
            cmd, data = pipe.recv()
            if cmd == "set_buffer":
                # shared-memory payload buffer: this worker owns row idx
                my_slice = data[0][idx]
                flags = data[1]
                pipe.send(("buffer_set", None))
            elif cmd == "reset":
                root_td = env.reset(data)
                if my_slice is not None:
                    _copy_into(my_slice.get("root_next"), root_td)
                    _signal_done()
                else:
                    pipe.send(("done", root_td))
            elif cmd == "step":
                root_td = root_td if root_td is not None else env.reset()
                if data is not None:
                    root_td.update(data)
                elif my_slice is not None:
                    for k in action_keys:
                        root_td.set(k, my_slice.get(k).clone())
                td = env.step(root_td)
                next_td = td.get("next")
                if my_slice is not None:
                    _copy_into(my_slice.get("next"), next_td)
                    _signal_done()
                else:
                    pipe.send(("done", next_td))
                root_td = step_mdp(td)
            elif cmd == "step_and_maybe_reset":
                root_td = root_td if root_td is not None else env.reset()
                if data is not None:
                    root_td.update(data)
                elif my_slice is not None:
                    for k in action_keys:
                        root_td.set(k, my_slice.get(k).clone())
                td, next_root = env.step_and_maybe_reset(root_td)
                if my_slice is not None:
                    _copy_into(my_slice.get("next"), td.get("next"))
                    _copy_into(my_slice.get("root_next"), next_root)
                    _signal_done()
                else:
                    pipe.send(("done", (td.get("next"), next_root)))
                root_td = next_root
            elif cmd == "seed":
                out = env.set_seed(data)
                pipe.send(("seeded", out))
            elif cmd == "state_dict":
                pipe.send(("state_dict", env.state_dict() if hasattr(env, "state_dict") else {}))
            elif cmd == "getattr":
                pipe.send(("attr", getattr(env, data)))
            elif cmd == "close":
                env.close()
                pipe.send(("closed", None))
                break
    except KeyboardInterrupt:
        pass
    except EOFError:
        pass
    except Exception:
        import traceback

        traceback.print_exc()
        raise


class ParallelEnv(BatchedEnvBase):
    """N worker processes with shared-memory payload (reference :1918)."""

    def __init__(self, num_workers: int, create_env_fn, **kwargs):
        super().__init__(num_workers, create_env_fn, **kwargs)
        ctx = mp.get_context("spawn")
        self.parent_pipes = []
        self.procs = []
        self._shared: Optional[TensorDictBase] = None
        # boot one worker first to learn specs, then allocate the shared
        # buffer, then boot the rest pointing at it
        metas = []
        # stage 1: spawn all workers WITHOUT buffers to collect specs
        for i in range(num_workers):
            parent, child = ctx.Pipe()
            proc = _ProcessNoWarn(
                target=_parallel_worker,
                args=(i, child, self.create_env_fn[i], self.create_env_kwargs[i], None),
            )
            proc.daemon = True
            proc.start()
            child.close()
            self.parent_pipes.append(parent)
            self.procs.append(proc)
        for pipe in self.parent_pipes:
            msg, meta = pipe.recv()
            assert msg == "meta"
            metas.append(meta)
        self._set_specs_from_meta(metas[0])
        self.is_closed = False
        self._root_cache: Optional[TensorDictBase] = None
        if self.shared_memory:
            self._setup_shared_buffer()

    def _setup_shared_buffer(self):
        """Allocate the shared-memory payload TensorDict (reference
        _create_td :1272, share :1414-1424): one [N]-row buffer holding
        action inputs, the step's ``next`` outputs and the post-reset
        ``root_next`` roots.  Workers write their row in place; pipes
        carry only 2-byte acks."""
        from ..tensordict import TensorDict as _TD

        n = self.num_workers
        buf = _TD({}, batch_size=[n])
        for k in self.full_action_spec.keys(True, True):
            buf.set(k, self.full_action_spec[k].zero())
        nxt = _TD({}, batch_size=[n])
        root = _TD({}, batch_size=[n])
        for k in self.full_observation_spec.keys(True, True):
            nxt.set(k, self.full_observation_spec[k].zero())
            root.set(k, self.full_observation_spec[k].zero())
        for k in self.full_reward_spec.keys(True, True):
            nxt.set(k, self.full_reward_spec[k].zero())
        for k in self.full_done_spec.keys(True, True):
            nxt.set(k, self.full_done_spec[k].zero())
            root.set(k, self.full_done_spec[k].zero())
        buf.set("next", nxt)
        buf.set("root_next", root)
        buf.share_memory_()
        self._shared = buf
        # completion flags: one shared int64 per worker; workers bump
        # their slot, the parent spin-waits the vector (reference
        # shm done-flags, batched_envs.py:3385-3397)
        self._flags = torch.zeros(n, dtype=torch.int64).share_memory_()
        self._seq = 0
        self._send_all("set_buffer", [(buf, self._flags)] * n)
        for pipe in self.parent_pipes:
            msg, _ = pipe.recv()
            assert msg == "buffer_set"
        if self.device is not None and self.device.type == "cuda":
            self._pin_shared_buffer()

    def _pin_shared_buffer(self):
        """Host-register the shared-memory pages (hipHostRegister via
        torch's cudart shim) so parent-side ``.to(device)`` copies are
        true async DMA from pinned memory (reference keeps pinned
        shared buffers for GPU parents)."""
        try:
            cudart = torch.cuda.cudart()
            for _k, v in self._shared.items(True, True):
                storage = v.untyped_storage()
                r = cudart.cudaHostRegister(storage.data_ptr(), storage.nbytes(), 0)
                if int(r) != 0:
                    raise RuntimeError(f"cudaHostRegister -> {r}")
            self._pinned = True
        except Exception as e:  # best-effort: unpinned copies still work
            logger.debug(f"ParallelEnv: host-register failed ({e!r})")
            self._pinned = False

    def _wait_flags(self):
        """Spin-wait until every worker's flag reaches the current
        sequence number (short pure spin, then yielding sleep)."""
        self._seq += 1
        seq = self._seq
        flags = self._flags
        deadline = time.monotonic() + BATCHED_PIPE_TIMEOUT
        spins = 0
        while True:
            if bool((flags >= seq).all()):
                return
            spins += 1
            if spins > 2000:
                time.sleep(5e-5)
            if spins % 512 == 0 and time.monotonic() > deadline:
                self._check()
                raise TimeoutError("ParallelEnv worker timed out (flags)")

    def _set_specs_from_meta(self, meta: EnvMetaData):
        n = self.num_workers
        self._inner_batch = meta.batch_size

        def ex(spec):
            return spec.expand(n, *spec.shape)

        self.full_observation_spec = ex(meta.specs["full_observation_spec"].clone())
        self.full_action_spec = ex(meta.specs["full_action_spec"].clone())
        self.full_reward_spec = ex(meta.specs["full_reward_spec"].clone())
        self.full_done_spec = ex(meta.specs["full_done_spec"].clone())

    def _check(self):
        _check_for_faulty_process(self.procs)

    def _send_all(self, cmd: str, datas):
        for pipe, data in zip(self.parent_pipes, datas):
            pipe.send((cmd, data))

    def _recv_all(self):
        outs = []
        for pipe in self.parent_pipes:
            if not pipe.poll(BATCHED_PIPE_TIMEOUT):
                self._check()
                raise TimeoutError("ParallelEnv worker timed out")
            msg, data = pipe.recv()
            outs.append(data)
        return outs

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        datas = []
        for i in range(self.num_workers):
            sub = tensordict[i].clone(False) if tensordict is not None else None
            datas.append(sub)
        self._send_all("reset", datas)
        if self._shared is not None:
            self._wait_flags()
            out = self._shared.get("root_next").clone()
        else:
            out = td_stack(self._recv_all(), 0)
        if self.device is not None and out.device != self.device:
            out = out.to(self.device, non_blocking=getattr(self, "_pinned", False))
        return out

    def _write_actions(self, tensordict: TensorDictBase) -> None:
        for k in self.full_action_spec.keys(True, True):
            val = tensordict.get(k, None)
            if val is not None:
                self._shared.get(k).copy_(val.cpu() if val.device.type != "cpu" else val)

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        input_keys = list(self.full_action_spec.keys(True, True))
        if self._shared is not None:
            self._write_actions(tensordict)
            self._send_all("step", [None] * self.num_workers)
            self._wait_flags()
            out = self._shared.get("next").clone()
        else:
            datas = []
            for i in range(self.num_workers):
                sub = tensordict[i].select(*input_keys, strict=False).cpu()
                datas.append(sub)
            self._send_all("step", datas)
            outs = self._recv_all()
            out = td_stack(outs, 0)
        if self.device is not None and out.device != self.device:
            out = out.to(self.device)
        return out

    def step_and_maybe_reset(self, tensordict: TensorDictBase):
        input_keys = list(self.full_action_spec.keys(True, True))
        if self._shared is not None:
            self._write_actions(tensordict)
            self._send_all("step_and_maybe_reset", [None] * self.num_workers)
            self._wait_flags()
            next_tds = self._shared.get("next").clone()
            next_roots = self._shared.get("root_next").clone()
        else:
            datas = []
            for i in range(self.num_workers):
                datas.append(tensordict[i].select(*input_keys, strict=False).cpu())
            self._send_all("step_and_maybe_reset", datas)
            outs = self._recv_all()
            next_tds = td_stack([o[0] for o in outs], 0)
            next_roots = td_stack([o[1] for o in outs], 0)
        if self.device is not None:
            nb = getattr(self, "_pinned", False)
            next_tds = next_tds.to(self.device, non_blocking=nb)
            next_roots = next_roots.to(self.device, non_blocking=nb)
        self._complete_done(next_tds)
        tensordict.set("next", next_tds)
        return tensordict, next_roots

    def _set_seed(self, seed: Optional[int]):
        out = seed
        for i, pipe in enumerate(self.parent_pipes):
            pipe.send(("seed", seed + i if seed is not None else None))
        for pipe in self.parent_pipes:
            msg, out = pipe.recv()
        return out

    def state_dict(self):
        self._send_all("state_dict", [None] * self.num_workers)
        outs = self._recv_all()
        return {f"worker{i}": sd for i, sd in enumerate(outs)}

    def close(self, raise_if_closed: bool = False):
        if self.is_closed:
            return
        for pipe in self.parent_pipes:
            try:
                pipe.send(("close", None))
            except (BrokenPipeError, OSError):
                pass
        for pipe in self.parent_pipes:
            try:
                if pipe.poll(5.0):
                    pipe.recv()
            except (EOFError, OSError):
                pass
        for proc in self.procs:
            proc.join(timeout=5.0)
            if proc.is_alive():
                proc.terminate()
        self.is_closed = True

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
