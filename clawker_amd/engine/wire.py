"""Framed-JSON wire protocol (Python side).

Mirror of native/common/util.hpp frame functions — keep in lockstep:
4-byte big-endian length, then UTF-8 JSON. Binary payloads are base64 in
a "data" field. Also provides SCM_RIGHTS fd passing.
"""
from __future__ import annotations

import array
import base64
import json
import socket
import struct
from typing import Any

MAX_FRAME = 16 * 1024 * 1024


class WireError(Exception):
    pass


def bind_unix(path, sock_type=socket.SOCK_STREAM) -> socket.socket:
    """Bind a unix socket, dodging the 108-byte sun_path limit via an
    O_PATH dirfd (/proc/self/fd/N/name)."""
    import os
    s = socket.socket(socket.AF_UNIX, sock_type)
    p = str(path)
    if len(p.encode()) < 100:
        s.bind(p)
        return s
    fd = os.open(os.path.dirname(p), os.O_PATH)
    try:
        s.bind(f"/proc/self/fd/{fd}/{os.path.basename(p)}")
    finally:
        os.close(fd)
    return s


def connect_unix(path, timeout: float | None = None) -> socket.socket:
    import os
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    if timeout is not None:
        s.settimeout(timeout)
    p = str(path)
    if len(p.encode()) < 100:
        s.connect(p)
        return s
    fd = os.open(os.path.dirname(p), os.O_PATH)
    try:
        s.connect(f"/proc/self/fd/{fd}/{os.path.basename(p)}")
    finally:
        os.close(fd)
    return s


def send_frame(sock: socket.socket, obj: dict[str, Any]) -> None:
    body = json.dumps(obj, separators=(",", ":")).encode()
    if len(body) > MAX_FRAME:
        raise WireError("frame too large")
    sock.sendall(struct.pack(">I", len(body)) + body)


def _recv_exact(sock: socket.socket, n: int) -> bytes | None:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            return None
        buf += chunk
    return buf


def recv_frame(sock: socket.socket) -> dict[str, Any] | None:
    hdr = _recv_exact(sock, 4)
    if hdr is None:
        return None
    (length,) = struct.unpack(">I", hdr)
    if length > MAX_FRAME:
        raise WireError("frame too large")
    body = _recv_exact(sock, length)
    if body is None:
        return None
    return json.loads(body)


def b64(data: bytes) -> str:
    return base64.b64encode(data).decode()


def unb64(s: str) -> bytes:
    return base64.b64decode(s)


def send_fd(sock: socket.socket, fd: int) -> None:
    sock.sendmsg([b"F"], [(socket.SOL_SOCKET, socket.SCM_RIGHTS, array.array("i", [fd]))])


def recv_fd(sock: socket.socket) -> int:
    msg, ancdata, _flags, _addr = sock.recvmsg(1, socket.CMSG_SPACE(4))
    if not msg:
        raise WireError("eof while receiving fd")
    for level, typ, data in ancdata:
        if level == socket.SOL_SOCKET and typ == socket.SCM_RIGHTS:
            fds = array.array("i")
            fds.frombytes(data[:4])
            return fds[0]
    raise WireError("no fd received")
