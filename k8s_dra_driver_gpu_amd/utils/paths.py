"""Filesystem path helpers."""

from __future__ import annotations

# Linux AF_UNIX sun_path limit (108 bytes including the NUL). gRPC fails
# with an opaque "Failed to add port" when exceeded; callers binding unix
# sockets check first so the operator sees the actual constraint.
AF_UNIX_PATH_MAX = 107


def check_unix_socket_path(path: str) -> str:
    if len(path.encode()) > AF_UNIX_PATH_MAX:
        raise ValueError(
            f"unix socket path exceeds the {AF_UNIX_PATH_MAX}-byte AF_UNIX "
            f"limit ({len(path.encode())} bytes): {path!r} — use a shorter "
            "--plugin-dir / registry dir"
        )
    return path
