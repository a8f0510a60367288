"""Dependency-free video writing for rollouts.

Reference capability: pytorch/rl torchrl/render/video.py (checkpoint →
video artifacts).  No video library ships in this image (no av/cv2/
PIL), so this module implements the GIF89a container + LZW directly:
``write_gif(frames, path, fps)`` turns a ``[T, C, H, W]`` (or
``[T, H, W, C]``) uint8/float tensor into an animated GIF with a
252-color uniform palette — enough for artifact inspection anywhere.
"""
from __future__ import annotations

import struct
from typing import Sequence, Union

import torch

__all__ = ["write_gif", "frames_to_uint8"]


def frames_to_uint8(frames: torch.Tensor) -> torch.Tensor:
    """Normalize to [T, H, W, 3] uint8."""
    f = frames.detach().cpu()
    if f.dim() == 3:  # [T, H, W] grayscale
        f = f.unsqueeze(-1).repeat(1, 1, 1, 3)
    if f.dim() != 4:
        raise ValueError(f"expected [T,C,H,W] or [T,H,W,C], got {tuple(f.shape)}")
    if f.shape[1] in (1, 3) and f.shape[-1] not in (1, 3):
        f = f.permute(0, 2, 3, 1)  # CHW → HWC
    if f.shape[-1] == 1:
        f = f.repeat(1, 1, 1, 3)
    if f.dtype.is_floating_point:
        f = (f.clamp(0, 1) * 255).to(torch.uint8)
    return f.contiguous()


def _lzw_encode(indices, code_size: int) -> bytes:
    """GIF-flavor LZW: emits clear/end codes, grows code width to 12."""
    clear = 1 << code_size
    end = clear + 1
    out = bytearray()
    cur = 0
    nbits = 0

    def emit(code, width):
        nonlocal cur, nbits
        cur |= code << nbits
        nbits += width
        while nbits >= 8:
            out.append(cur & 0xFF)
            cur >>= 8
            nbits -= 8

    table = {bytes([i]): i for i in range(clear)}
    next_code = end + 1
    width = code_size + 1
    emit(clear, width)
    buf = b""
    for idx in indices:
        nxt = buf + bytes([idx])
        if nxt in table:
            buf = nxt
        else:
            emit(table[buf], width)
            table[nxt] = next_code
            next_code += 1
            if next_code > (1 << width) and width < 12:
                width += 1
            elif next_code >= (1 << 12):
                emit(clear, width)
                table = {bytes([i]): i for i in range(clear)}
                next_code = end + 1
                width = code_size + 1
            buf = bytes([idx])
    if buf:
        emit(table[buf], width)
    emit(end, width)
    if nbits:
        out.append(cur & 0xFF)
    return bytes(out)


def write_gif(frames: torch.Tensor, path: str, fps: int = 30) -> str:
    """Write an animated GIF (infinite loop) from a frame tensor."""
    f = frames_to_uint8(frames)
    T, H, W, _ = f.shape
    # uniform 6x7x6 palette (252 colors)
    r = (f[..., 0].int() * 6) // 256
    g = (f[..., 1].int() * 7) // 256
    b = (f[..., 2].int() * 6) // 256
    idx = (r * 42 + g * 6 + b).to(torch.uint8)  # [T, H, W] in [0, 252)

    palette = bytearray()
    for ri in range(6):
        for gi in range(7):
            for bi in range(6):
                palette += bytes(
                    (
                        min(255, int((ri + 0.5) * 256 / 6)),
                        min(255, int((gi + 0.5) * 256 / 7)),
                        min(255, int((bi + 0.5) * 256 / 6)),
                    )
                )
    palette += b"\x00" * ((256 - 252) * 3)  # pad table to 256 entries

    delay_cs = max(1, round(100 / max(1, fps)))
    with open(path, "wb") as fh:
        fh.write(b"GIF89a")
        fh.write(struct.pack("<HHBBB", W, H, 0xF7, 0, 0))  # GCT, 256 colors
        fh.write(bytes(palette))
        # netscape loop extension
        fh.write(b"\x21\xff\x0bNETSCAPE2.0\x03\x01\x00\x00\x00")
        for t in range(T):
            fh.write(struct.pack("<BBBHBB", 0x21, 0xF9, 4, delay_cs, 0, 0))
            fh.write(struct.pack("<BHHHHB", 0x2C, 0, 0, W, H, 0))
            fh.write(bytes([8]))  # LZW min code size
            data = _lzw_encode(idx[t].reshape(-1).tolist(), 8)
            for off in range(0, len(data), 255):
                chunk = data[off : off + 255]
                fh.write(bytes([len(chunk)]) + chunk)
            fh.write(b"\x00")
        fh.write(b"\x3b")
    return path
