"""Response wrapper types.

Reference: pkg/gofr/http/response/raw.go:3-5 (Raw bypasses the JSON
envelope) and pkg/gofr/http/response/file.go:3-6 (File writes raw bytes
with the caller's Content-Type).
"""

from __future__ import annotations


class Raw:
    """Marshal `data` as JSON directly, without the {"data": ...} envelope."""

    __slots__ = ("data",)

    def __init__(self, data):
        self.data = data


class File:
    """Write raw bytes with the given Content-Type."""

    __slots__ = ("content", "content_type")

    def __init__(self, content: bytes, content_type: str):
        self.content = content
        self.content_type = content_type
