"""Image detection in action results.

Parity target: the reference detects image payloads in action results and
turns them into multimodal history entries, compressing via libvips
(reference: agent/image_detector.ex, utils/image_compressor.ex).  This
framework hosts text-only models (no vision pool member), so the MI355X
rebuild DETECTS images (magic bytes / data URLs), stores them as artifacts
on disk, and replaces the inline payload with a compact placeholder so huge
binary blobs never enter a model history.  Compression is a documented
divergence (no libvips in the image, no vision model to feed).
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


def sniff(data: bytes) -> Optional[str]:
    for magic, mime in _MAGIC:
        if data.startswith(magic):
            if mime == "image/webp" and data[8:12] != b"WEBP":
                continue
            return mime
    return None


def _store(data: bytes, mime: str) -> Dict[str, Any]:
    digest = hashlib.sha256(data).hexdigest()[:16]
    ext = mime.split("/")[-1]
    out_dir = os.environ.get(ARTIFACT_DIR_ENV) or "/tmp/quoracle_images"
    os.makedirs(out_dir, exist_ok=True)
    path = os.path.join(out_dir, f"{digest}.{ext}")
    if not os.path.exists(path):
        with open(path, "wb") as f:
            f.write(data)
    return {"mime": mime, "bytes": len(data), "sha256_16": digest,
            "path": path}


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
