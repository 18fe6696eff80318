"""Hand-written protobuf wire codec for the reference's tensor protocol.

Byte-compatible with /root/reference/shard/protos/mlx_tensor.proto
(package mlxtensor; messages Tensor/TensorResponse/ResetCache*) so a
reference driver can talk to our shard servers and vice versa.  Written
directly against the protobuf wire format because this image carries no
protoc/grpc_tools — and the messages are three fields each.

dtype strings: we EMIT the reference's MLX names ("mlx.core.float16")
— its bytes_to_tensor accepts only those and raises "Unsupported dtype"
otherwise (/root/reference/shard/utils.py:93-109) — and we ACCEPT both
MLX and torch spellings.  bf16 is first-class on our side (the
reference's numpy wire couldn't carry it); the wire_fp16 interop mode
downcasts to the reference-supported fp16 set.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

SERVICE = "mlxtensor.MLXTensorService"
SEND_TENSOR = f"/{SERVICE}/SendTensor"
RESET_CACHE = f"/{SERVICE}/ResetCache"

_DTYPE_FROM_STR = {
    "torch.float32": torch.float32,
    "torch.float16": torch.float16,
    "torch.bfloat16": torch.bfloat16,
    "torch.int32": torch.int32,
    "torch.int64": torch.int64,
    "mlx.core.float32": torch.float32,
    "mlx.core.float16": torch.float16,
    "mlx.core.bfloat16": torch.bfloat16,
    "mlx.core.int32": torch.int32,
    "mlx.core.int64": torch.int64,
    "float32": torch.float32,
    "float16": torch.float16,
    "bfloat16": torch.bfloat16,
    "int32": torch.int32,
    "int64": torch.int64,
}


_DTYPE_TO_STR = {
    torch.float32: "mlx.core.float32",
    torch.float16: "mlx.core.float16",
    torch.bfloat16: "mlx.core.bfloat16",
    torch.int32: "mlx.core.int32",
    torch.int64: "mlx.core.int64",
}


def dtype_to_str(dt: torch.dtype) -> str:
    """MLX spelling so a real reference peer can decode our messages
    (its map has no torch.* entries)."""
    try:
        return _DTYPE_TO_STR[dt]
    except KeyError:
        raise ValueError(f"unsupported wire dtype {dt}") from None


def dtype_from_str(s: str) -> torch.dtype:
    try:
        return _DTYPE_FROM_STR[s]
    except KeyError:
        raise ValueError(f"unsupported wire dtype {s!r}") from None


# --- varint ----------------------------------------------------------------

def _enc_varint(n: int, out: bytearray):
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _dec_varint(buf: memoryview, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7


def _enc_len_field(tag_byte: int, payload: bytes, out: bytearray):
    out.append(tag_byte)
    _enc_varint(len(payload), out)
    out += payload


# --- Tensor message --------------------------------------------------------

def encode_tensor(data: bytes, shape: List[int], dtype: str) -> bytes:
    out = bytearray()
    _enc_len_field(0x0A, data, out)                    # field 1: bytes
    if shape:
        packed = bytearray()
        for s in shape:
            _enc_varint(s, packed)
        _enc_len_field(0x12, bytes(packed), out)       # field 2: packed int32
    _enc_len_field(0x1A, dtype.encode(), out)          # field 3: string
    return bytes(out)


def decode_tensor(buf: bytes) -> Tuple[bytes, List[int], str]:
    mv = memoryview(buf)
    pos = 0
    data = b""
    shape: List[int] = []
    dtype = ""
    n = len(buf)
    while pos < n:
        tag, pos = _dec_varint(mv, pos)
        field, wt = tag >> 3, tag & 7
        if field == 1 and wt == 2:
            ln, pos = _dec_varint(mv, pos)
            data = bytes(mv[pos: pos + ln])
            pos += ln
        elif field == 2 and wt == 2:                    # packed
            ln, pos = _dec_varint(mv, pos)
            end = pos + ln
            while pos < end:
                v, pos = _dec_varint(mv, pos)
                shape.append(v)
        elif field == 2 and wt == 0:                    # unpacked
            v, pos = _dec_varint(mv, pos)
            shape.append(v)
        elif field == 3 and wt == 2:
            ln, pos = _dec_varint(mv, pos)
            dtype = bytes(mv[pos: pos + ln]).decode()
            pos += ln
        else:  # skip unknown
            if wt == 0:
                _, pos = _dec_varint(mv, pos)
            elif wt == 2:
                ln, pos = _dec_varint(mv, pos)
                pos += ln
            else:
                raise ValueError(f"unsupported wire type {wt}")
    return data, shape, dtype


# --- TensorResponse --------------------------------------------------------

def encode_tensor_response(success: bool, message: str = "",
                           tensor_msg: Optional[bytes] = None) -> bytes:
    out = bytearray()
    out.append(0x08)
    out.append(1 if success else 0)
    if message:
        _enc_len_field(0x12, message.encode(), out)
    if tensor_msg is not None:
        _enc_len_field(0x1A, tensor_msg, out)
    return bytes(out)


def decode_tensor_response(buf: bytes) -> Tuple[bool, str, Optional[bytes]]:
    mv = memoryview(buf)
    pos = 0
    success = False
    message = ""
    tensor_msg: Optional[bytes] = None
    n = len(buf)
    while pos < n:
        tag, pos = _dec_varint(mv, pos)
        field, wt = tag >> 3, tag & 7
        if field == 1 and wt == 0:
            v, pos = _dec_varint(mv, pos)
            success = bool(v)
        elif field == 2 and wt == 2:
            ln, pos = _dec_varint(mv, pos)
            message = bytes(mv[pos: pos + ln]).decode()
            pos += ln
        elif field == 3 and wt == 2:
            ln, pos = _dec_varint(mv, pos)
            tensor_msg = bytes(mv[pos: pos + ln])
            pos += ln
        else:
            if wt == 0:
                _, pos = _dec_varint(mv, pos)
            elif wt == 2:
                ln, pos = _dec_varint(mv, pos)
                pos += ln
            else:
                raise ValueError(f"unsupported wire type {wt}")
    return success, message, tensor_msg


# ResetCacheRequest is empty; ResetCacheResponse has the same two leading
# fields as TensorResponse.
def encode_reset_request() -> bytes:
    return b""


def encode_reset_response(success: bool, message: str = "") -> bytes:
    return encode_tensor_response(success, message, None)


def decode_reset_response(buf: bytes) -> Tuple[bool, str]:
    s, m, _ = decode_tensor_response(buf)
    return s, m


# --- torch <-> wire --------------------------------------------------------

def tensor_to_msg(t: torch.Tensor, wire_fp16: bool = False) -> bytes:
    """Serialize a torch tensor to a Tensor message.

    ``wire_fp16`` reproduces the reference's bf16→fp16 downcast
    (/root/reference/generate.py:69-70) for interop with reference peers
    whose numpy wire lacks bf16.
    """
    if wire_fp16 and t.dtype == torch.bfloat16:
        t = t.to(torch.float16)
    t = t.detach().contiguous().cpu()
    data = t.flatten().view(torch.uint8).numpy().tobytes()
    return encode_tensor(data, list(t.shape), dtype_to_str(t.dtype))


def msg_to_tensor(msg: bytes, device: str = "cpu") -> torch.Tensor:
    data, shape, dtype_s = decode_tensor(msg)
    dt = dtype_from_str(dtype_s)
    t = torch.frombuffer(bytearray(data), dtype=dt)
    if shape:
        t = t.reshape(shape)
    return t.to(device)
