"""Two-part wire codec for the request plane.

Framing (mirrors the reference's TwoPartCodec,
ai-dynamo/dynamo lib/runtime/src/pipeline/network/codec/two_part.rs:10-60):
24-byte prefix {header_len: u64, body_len: u64, xxh3_64(header+body): u64},
then header (JSON control metadata) then body (msgpack payload).
"""
from __future__ import annotations

import json
import struct
from typing import Any, Optional, Tuple

import msgpack
import xxhash

PREFIX = struct.Struct("<QQQ")
MAX_FRAME = 1 << 30
# thin native clients (capi/dynamo_client.hpp) may send this sentinel to
# skip checksum verification — TCP already guarantees integrity; xxh3 is
# only a corruption tripwire for file/relay transports
UNCHECKED = 0xFFFFFFFFFFFFFFFF


def encode_frame(header: dict, body: Any) -> bytes:
    h = json.dumps(header, separators=(",", ":")).encode()
    b = msgpack.packb(body, use_bin_type=True)
    csum = xxhash.xxh3_64_intdigest(h + b)
    return PREFIX.pack(len(h), len(b), csum) + h + b


def decode_prefix(buf: bytes) -> Tuple[int, int, int]:
    return PREFIX.unpack(buf)


def decode_frame(h: bytes, b: bytes, csum: int) -> Tuple[dict, Any]:
    if csum != UNCHECKED and xxhash.xxh3_64_intdigest(h + b) != csum:
        raise ValueError("frame checksum mismatch")
    return json.loads(h.decode()), msgpack.unpackb(b, raw=False)


async def read_frame(reader) -> Optional[Tuple[dict, Any]]:
    """Read one frame from an asyncio StreamReader; None on clean EOF."""
    try:
        prefix = await reader.readexactly(PREFIX.size)
    except (EOFError, ConnectionResetError, BrokenPipeError):
        return None
    except Exception as e:  # IncompleteReadError on EOF
        if type(e).__name__ == "IncompleteReadError":
            return None
        raise
    hlen, blen, csum = decode_prefix(prefix)
    if hlen > MAX_FRAME or blen > MAX_FRAME:
        raise ValueError("oversized frame")
    h = await reader.readexactly(hlen)
    b = await reader.readexactly(blen)
    return decode_frame(h, b, csum)
