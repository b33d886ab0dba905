"""Image detection in action results.

Parity target: the reference detects image payloads in action results and
turns them into multimodal history entries, compressing via libvips
(reference: agent/image_detector.ex, utils/image_compressor.ex).  This
framework hosts text-only models (no vision pool member), so the MI355X
rebuild DETECTS images (magic bytes / data URLs), stores them as artifacts
on disk, and replaces the inline payload with a compact placeholder so huge
binary blobs never enter a model history.  Oversized artifacts are
compressed on the way to disk (Pillow: bounded max dimension, JPEG quality
descent under a byte ceiling — the libvips-equivalent behavior); animated
GIFs and undecodable payloads are stored verbatim.
"""

from __future__ import annotations

import base64
import hashlib
import os
import re
from typing import Any, Dict, List, Optional, Tuple

_MAGIC = [
    (b"\x89PNG\r\n\x1a\n", "image/png"),
    (b"\xff\xd8\xff", "image/jpeg"),
    (b"GIF87a", "image/gif"),
    (b"GIF89a", "image/gif"),
    (b"RIFF", "image/webp"),       # + 'WEBP' at offset 8
    (b"BM", "image/bmp"),
]

_DATA_URL_RE = re.compile(
    r"data:(image/[a-z+.-]+);base64,([A-Za-z0-9+/=]{64,})")

ARTIFACT_DIR_ENV = "QUORACLE_IMAGE_DIR"
MAX_DIM = 2048               # longest side after compression
MAX_BYTES = 5 * 1024 * 1024  # byte ceiling before compression kicks in


def compress_image(data: bytes, mime: str,
                   max_dim: int = MAX_DIM,
                   max_bytes: int = MAX_BYTES) -> Tuple[bytes, str]:
    """Bound an image's size (reference: utils/image_compressor.ex via
    libvips; here Pillow): downscale so the longest side <= max_dim, then
    JPEG-re-encode with descending quality until under max_bytes.  Returns
    (data, mime) — unchanged when already small enough, undecodable, or
    animated."""
    if len(data) <= max_bytes:
        return data, mime
    try:
        import io
        from PIL import Image
        img = Image.open(io.BytesIO(data))
        if getattr(img, "is_animated", False):
            return data, mime
        img.load()
    except Exception:  # noqa: BLE001 — store verbatim if not decodable
        return data, mime
    if max(img.size) > max_dim:
        img.thumbnail((max_dim, max_dim))
    if img.mode not in ("RGB", "L"):
        img = img.convert("RGB")
    best = data
    for quality in (85, 70, 55, 40, 30):
        buf = io.BytesIO()
        img.save(buf, format="JPEG", quality=quality)
        best = buf.getvalue()
        if len(best) <= max_bytes:
            break
    if len(best) >= len(data):
        return data, mime
    return best, "image/jpeg"


def sniff(data: bytes) -> Optional[str]:
    for magic, mime in _MAGIC:
        if data.startswith(magic):
            if mime == "image/webp" and data[8:12] != b"WEBP":
                continue
            return mime
    return None


def _store(data: bytes, mime: str) -> Dict[str, Any]:
    original_bytes = len(data)
    data, mime = compress_image(data, mime)
    digest = hashlib.sha256(data).hexdigest()[:16]
    ext = mime.split("/")[-1]
    out_dir = os.environ.get(ARTIFACT_DIR_ENV) or "/tmp/quoracle_images"
    os.makedirs(out_dir, exist_ok=True)
    path = os.path.join(out_dir, f"{digest}.{ext}")
    if not os.path.exists(path):
        with open(path, "wb") as f:
            f.write(data)
    return {"mime": mime, "bytes": len(data), "sha256_16": digest,
            "path": path, "original_bytes": original_bytes}


def extract_images(result: Any) -> Tuple[Any, List[Dict[str, Any]]]:
    """Recursively find image payloads in an action result.

    Returns (result_with_placeholders, artifacts).  Detected forms:
    raw bytes values with image magic, and base64 data-URLs inside strings.
    """
    artifacts: List[Dict[str, Any]] = []

    def _walk(value: Any) -> Any:
        if isinstance(value, bytes):
            mime = sniff(value)
            if mime:
                art = _store(value, mime)
                artifacts.append(art)
                return (f"[image artifact {art['sha256_16']} "
                        f"{mime} {art['bytes']}B -> {art['path']}]")
            return value
        if isinstance(value, str):
            def _sub(m: re.Match) -> str:
                try:
                    raw = base64.b64decode(m.group(2), validate=False)
                except Exception:  # noqa: BLE001
                    return m.group(0)
                art = _store(raw, m.group(1))
                artifacts.append(art)
                return (f"[image artifact {art['sha256_16']} "
                        f"{m.group(1)} {art['bytes']}B -> {art['path']}]")
            return _DATA_URL_RE.sub(_sub, value)
        if isinstance(value, dict):
            return {k: _walk(v) for k, v in value.items()}
        if isinstance(value, list):
            return [_walk(v) for v in value]
        return value

    return _walk(result), artifacts
