"""Length-prefixed message framing over TCP (SURVEY.md §2 "Socket RPC framing").

Frame layout: [u32 body_len][u8 kind_len][kind: ascii][payload bytes].
Used by the control plane (DHT façade RPC, cross-host expert RPC fallback,
averager part exchange). The intra-node tensor plane never uses sockets —
it is RCCL all-to-all over xGMI (learning_at_home_amd/parallel).
"""
from __future__ import annotations

import asyncio
import socket
import struct
from typing import Tuple

Endpoint = Tuple[str, int]  # (host, port)

_LEN = struct.Struct(">I")
MAX_MESSAGE_BYTES = 1 << 30


def send_message(sock: socket.socket, kind: str, payload: bytes = b"") -> None:
    kind_b = kind.encode("ascii")
    assert len(kind_b) < 256
    body = bytes([len(kind_b)]) + kind_b + payload
    sock.sendall(_LEN.pack(len(body)) + body)


def _recv_exactly(sock: socket.socket, n: int) -> bytes:
    chunks = []
    remaining = n
    while remaining > 0:
        chunk = sock.recv(min(remaining, 1 << 20))
        if not chunk:
            raise ConnectionError("socket closed mid-message")
        chunks.append(chunk)
        remaining -= len(chunk)
    return b"".join(chunks)


def recv_message(sock: socket.socket) -> Tuple[str, bytes]:
    (body_len,) = _LEN.unpack(_recv_exactly(sock, 4))
    if body_len > MAX_MESSAGE_BYTES:
        raise ValueError(f"message too large: {body_len}")
    body = _recv_exactly(sock, body_len)
    kind_len = body[0]
    kind = body[1 : 1 + kind_len].decode("ascii")
    return kind, body[1 + kind_len :]


async def awrite_message(writer: asyncio.StreamWriter, kind: str, payload: bytes = b"") -> None:
    kind_b = kind.encode("ascii")
    assert len(kind_b) < 256
    body = bytes([len(kind_b)]) + kind_b + payload
    writer.write(_LEN.pack(len(body)) + body)
    await writer.drain()


async def aread_message(reader: asyncio.StreamReader) -> Tuple[str, bytes]:
    header = await reader.readexactly(4)
    (body_len,) = _LEN.unpack(header)
    if body_len > MAX_MESSAGE_BYTES:
        raise ValueError(f"message too large: {body_len}")
    body = await reader.readexactly(body_len)
    kind_len = body[0]
    kind = body[1 : 1 + kind_len].decode("ascii")
    return kind, body[1 + kind_len :]
