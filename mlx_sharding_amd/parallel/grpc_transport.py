"""gRPC transport: stage client + server plumbing.

Method paths and message bytes match the reference's generated stubs
(/root/reference/shard/grpc/mlx_tensor_pb2_grpc.py:33-86) via the
hand-written codec in wire.py — no protoc needed.  Used for the CPU
plumbing config and as the control plane; the GPU data path is RCCL
(parallel/rccl.py).
"""

from __future__ import annotations

import threading
import time
from concurrent import futures
from typing import Callable, List, Sequence

import grpc
import torch

from . import wire

MAX_MSG = 1280 * 1024 * 1024  # reference server limit (/root/reference/shard/server/server.py:78-82)

_CHANNEL_OPTS = [
    ("grpc.max_send_message_length", MAX_MSG),
    ("grpc.max_receive_message_length", MAX_MSG),
    ("grpc.max_metadata_size", 32 * 1024 * 1024),
]


class ShardUnavailable(RuntimeError):
    """A stage server could not be reached (dead/refused/timeout).

    The reference surfaces this as a swallowed ``None`` + crash on the
    next op (/root/reference/shard/utils.py:79-85); the build raises a
    typed error the API layer turns into a clean 502."""


class StageClient:
    """Client stub for one remote shard server.

    ``retries`` bounded re-attempts on transient UNAVAILABLE (server
    restarting); anything else fails fast with ShardUnavailable."""

    def __init__(self, address: str, retries: int = 1,
                 timeout_s: float = 300.0):
        self.address = address
        self.retries = retries
        self.timeout_s = timeout_s
        self._channel = grpc.insecure_channel(address, options=_CHANNEL_OPTS)
        self._send = self._channel.unary_unary(wire.SEND_TENSOR)
        self._reset = self._channel.unary_unary(wire.RESET_CACHE)

    def _call(self, fn, payload: bytes) -> bytes:
        last = None
        for attempt in range(self.retries + 1):
            try:
                return fn(payload, timeout=self.timeout_s)
            except grpc.RpcError as e:
                last = e
                code = e.code() if hasattr(e, "code") else None
                if code != grpc.StatusCode.UNAVAILABLE or attempt == self.retries:
                    break
                time.sleep(0.2 * (attempt + 1))
        code = last.code().name if hasattr(last, "code") else "RPC_ERROR"
        raise ShardUnavailable(
            f"shard {self.address} unreachable ({code}): "
            f"{last.details() if hasattr(last, 'details') else last}") from None

    def send_tensor(self, t: torch.Tensor, wire_fp16: bool = False,
                    device: str = "cpu") -> torch.Tensor:
        resp = self._call(self._send, wire.tensor_to_msg(t, wire_fp16=wire_fp16))
        ok, message, tmsg = wire.decode_tensor_response(bytes(resp))
        if not ok or tmsg is None:
            raise RuntimeError(f"shard {self.address} SendTensor failed: {message}")
        return wire.msg_to_tensor(tmsg, device=device)

    def reset_cache(self):
        resp = self._call(self._reset, wire.encode_reset_request())
        ok, message = wire.decode_reset_response(bytes(resp))
        if not ok:
            raise RuntimeError(f"shard {self.address} ResetCache failed: {message}")

    def healthy(self) -> bool:
        """Cheap liveness probe: ResetCache on a fresh session is a no-op
        for correctness (every generation resets first) and doubles as a
        health check the reference never had."""
        try:
            self.reset_cache()
            return True
        except (RuntimeError, grpc.RpcError):
            return False

    def close(self):
        self._channel.close()


def make_clients(addresses: Sequence[str]) -> List[StageClient]:
    return [StageClient(a.strip()) for a in addresses if a.strip()]


def serve_forward(forward: Callable[[torch.Tensor], torch.Tensor],
                  reset: Callable[[], None],
                  port: int = 0,
                  max_workers: int = 10) -> "grpc.Server":
    """Start a gRPC server exposing SendTensor/ResetCache around the given
    forward/reset callables.  Returns the started server; the bound port is
    ``server._mlxs_port`` (auto-assigned when port=0, like the reference's
    '[::]:0' + printed port, server.py:88-90)."""
    lock = threading.Lock()

    def handle_send(request: bytes, context) -> bytes:
        try:
            t = wire.msg_to_tensor(bytes(request))
            with lock:  # serialize forwards; the KV cache is stateful
                out = forward(t)
            return wire.encode_tensor_response(
                True, "", wire.tensor_to_msg(out))
        except Exception as e:  # noqa: BLE001
            return wire.encode_tensor_response(False, f"{type(e).__name__}: {e}")

    def handle_reset(request: bytes, context) -> bytes:
        try:
            with lock:
                reset()
            return wire.encode_reset_response(True, "Cache reset successfully")
        except Exception as e:  # noqa: BLE001
            return wire.encode_reset_response(False, f"{type(e).__name__}: {e}")

    handlers = grpc.method_handlers_generic_handler(
        wire.SERVICE,
        {
            "SendTensor": grpc.unary_unary_rpc_method_handler(handle_send),
            "ResetCache": grpc.unary_unary_rpc_method_handler(handle_reset),
        },
    )
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers),
                         options=_CHANNEL_OPTS)
    server.add_generic_rpc_handlers((handlers,))
    bound = server.add_insecure_port(f"[::]:{port}")
    server._mlxs_port = bound
    server.start()
    return server
