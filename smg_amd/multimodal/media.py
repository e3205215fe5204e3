"""Media fetch + decode (reference: crates/multimodal/src/media.rs (2,361 LoC)
— image fetch from data:/base64/http(s)/file, decode, EXIF-free RGB).

Decode via PIL (the accuracy reference); resize+normalize then runs on the
gfx950 kernel (csrc/image.hip)."""
from __future__ import annotations

import base64
import binascii
import io
import numpy as np


class MediaError(ValueError):
    pass


async def fetch_image_bytes(url: str, session=None, max_bytes: int = 64 << 20) -> bytes:
    if url.startswith("data:"):
        try:
            header, payload = url.split(",", 1)
        except ValueError:
            raise MediaError("malformed data: URL")
        if ";base64" in header:
            try:
                return base64.b64decode(payload)
            except binascii.Error as e:
                raise MediaError(f"invalid base64 payload: {e}")
        return payload.encode()
    if url.startswith(("http://", "https://")):
        if session is None:
            raise MediaError("http image fetch requires a client session")
        async with session.get(url) as resp:
            if resp.status != 200:
                raise MediaError(f"image fetch failed: HTTP {resp.status}")
            data = await resp.read()
            if len(data) > max_bytes:
                raise MediaError("image too large")
            return data
    if url.startswith("file://"):
        with open(url[7:], "rb") as f:
            return f.read()
    raise MediaError(f"unsupported image URL scheme: {url[:32]}")


def decode_image(data: bytes) -> np.ndarray:
    """-> u8 HWC RGB array."""
    from PIL import Image

    try:
        img = Image.open(io.BytesIO(data))
        img = img.convert("RGB")
    except Exception as e:
        raise MediaError(f"image decode failed: {e}")
    return np.asarray(img)
