"""Multimodal input detection and message assembly (reference parity:
multimodal.py / multimodal_responses.py — input type classification, data-URL
handling, and OpenAI-style typed content parts).

The shipped Llama configs are text-only, so AgentAI raises
UnsupportedModality when a non-text part reaches a text-only runner — but
the detection/assembly layer is model-independent and fully implemented so
a multimodal model family can plug in without SDK changes.
"""
from __future__ import annotations

import base64
import os
import re

_IMAGE_EXT = {".png", ".jpg", ".jpeg", ".gif", ".webp", ".bmp"}
_AUDIO_EXT = {".wav", ".mp3", ".flac", ".ogg", ".m4a", ".opus"}
_DATA_URL = re.compile(r"^data:(?P<mime>[\w.+-]+/[\w.+-]+);base64,(?P<b64>.+)$",
                       re.DOTALL)

# magic-byte sniffing for raw bytes inputs
_MAGIC = [
    (b"\x89PNG\r\n\x1a\n", "image/png"),
    (b"\xff\xd8\xff", "image/jpeg"),
    (b"GIF8", "image/gif"),
    (b"RIFF", None),  # container: WAV or WEBP, resolved below
    (b"ID3", "audio/mpeg"),
    (b"\xff\xfb", "audio/mpeg"),
    (b"fLaC", "audio/flac"),
    (b"OggS", "audio/ogg"),
]


class UnsupportedModality(TypeError):
    """Raised when a non-text content part reaches a text-only model."""


def sniff_bytes(data: bytes) -> str | None:
    """MIME type from magic bytes, or None if unrecognized."""
    for magic, mime in _MAGIC:
        if data.startswith(magic):
            if mime is not None:
                return mime
            if data[8:12] == b"WAVE":
                return "audio/wav"
            if data[8:12] == b"WEBP":
                return "image/webp"
    return None


def detect_part(obj) -> dict:
    """Classify one ai() input into a typed content part:
      {"type": "text", "text": ...}
      {"type": "image_url", "image_url": {"url": "data:...;base64,..."}}
      {"type": "input_audio", "input_audio": {"data": b64, "format": ...}}
    Accepts: str (plain text, data: URL, http(s) URL, or an existing local
    image/audio file path), bytes (sniffed), or an already-typed dict.
    """
    if isinstance(obj, dict) and "type" in obj:
        return obj
    if isinstance(obj, (bytes, bytearray)):
        mime = sniff_bytes(bytes(obj[:16])) or "application/octet-stream"
        return _part_from_mime(mime, base64.b64encode(bytes(obj)).decode())
    s = str(obj)
    m = _DATA_URL.match(s)
    if m:
        return _part_from_mime(m.group("mime"), m.group("b64"), url=s)
    if s.startswith(("http://", "https://")):
        ext = os.path.splitext(s.split("?", 1)[0])[1].lower()
        if ext in _IMAGE_EXT:
            return {"type": "image_url", "image_url": {"url": s}}
        if ext in _AUDIO_EXT:
            return {"type": "audio_url", "audio_url": {"url": s}}
        return {"type": "text", "text": s}
    ext = os.path.splitext(s)[1].lower()
    if (ext in _IMAGE_EXT or ext in _AUDIO_EXT) and os.path.isfile(s):
        with open(s, "rb") as f:
            data = f.read()
        mime = sniff_bytes(data[:16]) or \
            ("image/" + ext[1:] if ext in _IMAGE_EXT else "audio/" + ext[1:])
        return _part_from_mime(mime, base64.b64encode(data).decode())
    return {"type": "text", "text": s}


def _part_from_mime(mime: str, b64: str, url: str | None = None) -> dict:
    if mime.startswith("image/"):
        return {"type": "image_url",
                "image_url": {"url": url or f"data:{mime};base64,{b64}"}}
    if mime.startswith("audio/"):
        return {"type": "input_audio",
                "input_audio": {"data": b64, "format": mime.split("/", 1)[1]}}
    return {"type": "text", "text": b64 if url is None else url}


def build_content(parts) -> list[dict] | str:
    """Assemble typed parts from mixed inputs.  Returns a plain string when
    everything is text (the cheap common case), else the typed list."""
    typed = [detect_part(p) for p in parts]
    if all(p.get("type") == "text" for p in typed):
        return "\n".join(p["text"] for p in typed)
    return typed


def require_text(content, model: str) -> str:
    """Flatten content for a text-only model; raise on non-text parts."""
    if isinstance(content, str):
        return content
    bad = [p["type"] for p in content if p.get("type") != "text"]
    if bad:
        raise UnsupportedModality(
            f"model '{model}' is text-only but got {bad} content; "
            "serve a multimodal model family to use ai_with_vision/audio")
    return "\n".join(p["text"] for p in content)
