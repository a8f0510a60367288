"""Lazy video-clip references (reference torchrl/data/video.py:278).

A :class:`VideoClipRef` stores only frame ADDRESSES — source file ids
and per-frame indices — never decoded pixels or open decoders, so it is
picklable and indexes like a tensor of frames.  Decoding (torchvision/
av) is gated and happens only on an explicit ``decode()`` call.
"""
from __future__ import annotations

import importlib.util
from typing import List, Sequence

import torch

__all__ = ["VideoClipRef"]


class VideoClipRef:
    def __init__(self, sources: Sequence[str], frame_index: torch.Tensor,
                 file_id: torch.Tensor):
        self.sources = list(sources)
        self.frame_index = torch.as_tensor(frame_index, dtype=torch.long)
        self.file_id = torch.as_tensor(file_id, dtype=torch.long)
        if self.frame_index.shape != self.file_id.shape:
            raise ValueError("frame_index and file_id must align")

    @property
    def batch_size(self):
        return self.frame_index.shape

    def __len__(self):
        return self.frame_index.shape[0]

    def __getitem__(self, idx) -> "VideoClipRef":
        return VideoClipRef(self.sources, self.frame_index[idx], self.file_id[idx])

    def __repr__(self):
        return (f"VideoClipRef(frames={tuple(self.frame_index.shape)}, "
                f"sources={len(self.sources)})")

    def decode(self) -> torch.Tensor:
        if importlib.util.find_spec("torchvision") is None:
            raise ImportError(
                "VideoClipRef.decode requires torchvision, which is not "
                "installed in this image."
            )
        raise NotImplementedError("video decoding scaffolding")
