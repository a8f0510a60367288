"""LLM weight publication: trainer → generation-engine weight transfer.

Reference capabilities rebuilt MI355X-first:
* packed collective broadcast — pytorch/rl
  torchrl/weight_update/llm/vllm_nccl.py:356 (``_pack_weights`` +
  NCCL broadcast :405, metadata ``get_model_metadata``): on ROCm this is
  ONE RCCL broadcast over xGMI of a single contiguous buffer holding
  every parameter, from the trainer rank to all generation workers.
* memmap double-buffer — torchrl/weight_update/llm/vllm_double_buffer.py:149:
  no collective at all; the trainer writes alternating A/B flat binary
  buffers on shared storage plus an atomic version file, readers poll
  the version and map the finished buffer.  Survives trainer/worker
  restarts and crosses node boundaries without a process group.

Both schemes speak plain ``nn.Module`` weights, so they drive the
transformers-backed generation wrapper
(:class:`rl_amd.modules.llm.TransformersWrapper`) directly, and the same
metadata/packing contract applies to a vLLM/SGLang engine worker when
those engines are present.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Optional, Tuple

import torch

from ..tensordict import TensorDict
from .weight_sync_schemes import WeightStrategy, WeightSyncScheme

__all__ = [
    "get_model_metadata",
    "LLMCollectiveWeightSyncScheme",
    "LLMDoubleBufferWeightSyncScheme",
]


def get_model_metadata(model) -> Dict[str, Tuple[str, Tuple[int, ...]]]:
    """Name → (dtype, shape) for every parameter+buffer, in the fixed
    iteration order the packed transfer uses (reference
    vllm_nccl.py ``get_model_metadata``)."""
    sd = model.state_dict() if hasattr(model, "state_dict") else dict(model)
    return {
        k: (str(v.dtype).replace("torch.", ""), tuple(v.shape))
        for k, v in sd.items()
        if isinstance(v, torch.Tensor)
    }


def _pack(sd: Dict[str, torch.Tensor], device) -> torch.Tensor:
    """One contiguous uint8 buffer holding every tensor back-to-back."""
    total = sum(v.numel() * v.element_size() for v in sd.values())
    buf = torch.empty(total, dtype=torch.uint8, device=device)
    off = 0
    for v in sd.values():
        n = v.numel() * v.element_size()
        buf[off : off + n] = (
            v.detach().contiguous().view(-1).view(torch.uint8).to(device)
        )
        off += n
    return buf

def _unpack_into(buf: torch.Tensor, sd: Dict[str, torch.Tensor]) -> None:
    off = 0
    with torch.no_grad():
        for v in sd.values():
            n = v.numel() * v.element_size()
            src = buf[off : off + n].to(v.device).view(v.dtype).view(v.shape)
            v.copy_(src)
            off += n


class LLMCollectiveWeightSyncScheme(WeightSyncScheme):
    """Packed-buffer collective broadcast of a (large) model's weights.

    Trainer (``src`` rank) and every generation worker join the same
    process group ("nccl" = RCCL on ROCm for GPU weights, gloo for CPU);
    ``send()`` packs all parameters into one contiguous buffer and
    issues a single broadcast — for GPT-2-class models that is one
    ~0.5 GB RCCL transfer saturating the xGMI link instead of hundreds
    of per-tensor collectives.  ``receive()`` participates in the same
    broadcast and scatters the buffer back into the local module.

    Reference: vllm_nccl.py:356 (packed broadcast), :405 (scheme).
    """

    def __init__(self, src: int = 0, group=None, device=None):
        super().__init__("state_dict")
        self.src = src
        self.group = group
        self.device = device
        self._version = 0

    def _comm_device(self, sd):
        if self.device is not None:
            return torch.device(self.device)
        import torch.distributed as dist

        backend = dist.get_backend(self.group)
        if backend == "nccl" and torch.cuda.is_available():
            return torch.device("cuda", torch.cuda.current_device())
        return torch.device("cpu")

    def send(self, weights=None) -> None:
        import torch.distributed as dist

        model = weights if weights is not None else self.model
        sd = {
            k: v
            for k, v in model.state_dict().items()
            if isinstance(v, torch.Tensor)
        }
        dev = self._comm_device(sd)
        buf = _pack(sd, dev)
        dist.broadcast(buf, src=self.src, group=self.group)
        self._version += 1

    def receive(self, model) -> bool:
        import torch.distributed as dist

        sd = {
            k: v
            for k, v in model.state_dict().items()
            if isinstance(v, torch.Tensor)
        }
        dev = self._comm_device(sd)
        total = sum(v.numel() * v.element_size() for v in sd.values())
        buf = torch.empty(total, dtype=torch.uint8, device=dev)
        dist.broadcast(buf, src=self.src, group=self.group)
        _unpack_into(buf, sd)
        self._version += 1
        return True


class LLMDoubleBufferWeightSyncScheme(WeightSyncScheme):
    """Memmap double-buffer weight transfer (no collective).

    The trainer alternates between two flat binary buffers under
    ``path`` (``weights_a.bin`` / ``weights_b.bin``), writes the inactive
    one, then atomically bumps ``version.json`` naming the finished
    buffer.  Readers poll the version file and ``np.memmap`` the named
    buffer straight into their module — a reader can never observe a
    half-written buffer, the trainer never blocks on readers, and the
    scheme crosses node boundaries over any shared filesystem.

    Reference: vllm_double_buffer.py:149.
    """

    def __init__(self, path: str, model=None):
        super().__init__("state_dict")
        self.path = path
        os.makedirs(path, exist_ok=True)
        self._next = "a"
        self._seen_version = -1
        if model is not None:
            self.connect(model)

    # -- trainer side --------------------------------------------------- #
    def _meta_path(self):
        return os.path.join(self.path, "metadata.json")

    def send(self, weights=None) -> None:
        model = weights if weights is not None else self.model
        sd = {
            k: v
            for k, v in model.state_dict().items()
            if isinstance(v, torch.Tensor)
        }
        meta = get_model_metadata(model)
        if not os.path.exists(self._meta_path()):
            tmp = self._meta_path() + ".tmp"
            with open(tmp, "w") as f:
                json.dump(meta, f)
            os.replace(tmp, self._meta_path())
        buf = _pack(sd, torch.device("cpu"))
        name = f"weights_{self._next}.bin"
        target = os.path.join(self.path, name)
        tmp = target + ".tmp"
        buf.numpy().tofile(tmp)
        os.replace(tmp, target)
        version = self._read_version()[0] + 1
        vtmp = os.path.join(self.path, "version.json.tmp")
        with open(vtmp, "w") as f:
            json.dump({"version": version, "buffer": name}, f)
        os.replace(vtmp, os.path.join(self.path, "version.json"))
        self._next = "b" if self._next == "a" else "a"

    # -- worker side ---------------------------------------------------- #
    def _read_version(self):
        try:
            with open(os.path.join(self.path, "version.json")) as f:
                d = json.load(f)
            return d["version"], d["buffer"]
        except (OSError, ValueError, KeyError):
            return 0, None

    def receive(self, model) -> bool:
        import numpy as np

        version, name = self._read_version()
        if name is None or version == self._seen_version:
            return False
        arr = np.memmap(os.path.join(self.path, name), dtype=np.uint8, mode="r")
        # the reader only COPIES from this view (torch warns about
        # non-writable arrays; writing would indeed be UB — we never do)
        import warnings

        with warnings.catch_warnings():
            warnings.simplefilter("ignore", UserWarning)
            buf = torch.from_numpy(np.asarray(arr))
        sd = {
            k: v
            for k, v in model.state_dict().items()
            if isinstance(v, torch.Tensor)
        }
        _unpack_into(buf, sd)
        self._seen_version = version
        return True
