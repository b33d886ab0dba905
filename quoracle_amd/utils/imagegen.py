"""Local procedural image model — the generate_images backing when no
external image model is injected.

The reference generates images through a configured provider model
(reference: lib/quoracle/models/image_query.ex,
actions/generate_images.ex).  This environment has no network and no
diffusion checkpoint, so the locally-hosted equivalent is a deterministic
procedural renderer: the prompt (plus optional source image bytes for
edit mode) seeds a small spectral-noise field that is rendered to a real
PNG (pure numpy + zlib, no imaging libraries needed).  It is a genuine
end-to-end image path — real bytes, real artifacts on disk, multimodal
history entries — with the renderer standing in for a diffusion model
the way random-init weights stand in for trained checkpoints everywhere
else in the benchmark environment.
"""

from __future__ import annotations

import hashlib
import struct
import zlib
from typing import Optional, Tuple

import numpy as np

MODEL_NAME = "local-procedural-v0"


def _png_chunk(tag: bytes, payload: bytes) -> bytes:
    return (struct.pack(">I", len(payload)) + tag + payload
            + struct.pack(">I", zlib.crc32(tag + payload) & 0xFFFFFFFF))


def encode_png(rgb: np.ndarray) -> bytes:
    """Encode an HxWx3 uint8 array as a PNG (filter 0 rows, zlib)."""
    h, w, _ = rgb.shape
    raw = b"".join(b"\x00" + rgb[y].tobytes() for y in range(h))
    ihdr = struct.pack(">IIBBBBB", w, h, 8, 2, 0, 0, 0)
    return (b"\x89PNG\r\n\x1a\n"
            + _png_chunk(b"IHDR", ihdr)
            + _png_chunk(b"IDAT", zlib.compress(raw, 6))
            + _png_chunk(b"IEND", b""))


def render(prompt: str, size: Tuple[int, int] = (256, 256),
           source_image: Optional[bytes] = None) -> bytes:
    """Render a deterministic PNG for the prompt (edit mode perturbs the
    seed with the source image's digest)."""
    seed_src = prompt.encode("utf-8", "replace")
    if source_image:
        seed_src += hashlib.sha256(source_image).digest()
    seed = int.from_bytes(hashlib.sha256(seed_src).digest()[:8], "big")
    rng = np.random.default_rng(seed)
    h, w = size[1], size[0]
    yy, xx = np.mgrid[0:h, 0:w].astype(np.float32)
    yy /= h
    xx /= w
    field = np.zeros((h, w), dtype=np.float32)
    for _ in range(6):                      # spectral noise octaves
        fx, fy = rng.uniform(1.0, 9.0, 2)
        phase = rng.uniform(0, 2 * np.pi)
        amp = rng.uniform(0.3, 1.0)
        field += amp * np.sin(2 * np.pi * (fx * xx + fy * yy) + phase)
    field = (field - field.min()) / max(1e-6, field.max() - field.min())
    # palette: three anchor colors from the seed, interpolated by field
    anchors = rng.integers(0, 256, (3, 3)).astype(np.float32)
    t = field[..., None]
    rgb = (anchors[0] * (1 - t) ** 2 + anchors[1] * 2 * t * (1 - t)
           + anchors[2] * t ** 2)
    return encode_png(np.clip(rgb, 0, 255).astype(np.uint8))
