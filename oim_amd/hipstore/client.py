"""JSON-RPC 2.0 client over a Unix socket (reference pkg/spdk/client.go).

Framing is SPDK's: concatenated JSON objects in both directions, no
length prefix.  Error behavior matches the reference: RPC errors raise
:class:`RpcError` carrying the JSON-RPC code; SPDK-style "not found"
often arrives as ERROR_INVALID_PARAMS (-32602) and callers probe for
that (reference client.go:60-85, local.go:53-57).
"""

from __future__ import annotations

import json
import socket
import threading
from typing import Any, Optional

from ..log import from_context

ERROR_PARSE_ERROR = -32700
ERROR_INVALID_REQUEST = -32600
ERROR_METHOD_NOT_FOUND = -32601
ERROR_INVALID_PARAMS = -32602
ERROR_INTERNAL = -32603


class RpcError(Exception):
    def __init__(self, code: int, message: str):
        super().__init__(f"code: {code} msg: {message}")
        self.code = code
        self.message = message

    def is_not_found(self) -> bool:
        """SPDK reports 'not found' as invalid-params; tolerate both."""
        return self.code in (ERROR_INVALID_PARAMS, ERROR_METHOD_NOT_FOUND)


class Client:
    """One connection; safe for concurrent invoke() via an internal lock."""

    def __init__(self, socket_path: str, timeout: float = 60.0):
        self._path = socket_path
        self._sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self._sock.settimeout(timeout)
        self._sock.connect(socket_path)
        self._buffer = b""
        self._next_id = 1
        self._lock = threading.Lock()
        self._decoder = json.JSONDecoder()

    def close(self) -> None:
        try:
            self._sock.close()
        except OSError:
            pass

    def __enter__(self) -> "Client":
        return self

    def __exit__(self, *exc) -> None:
        self.close()

    def invoke(self, method: str, params: Optional[dict] = None) -> Any:
        """Send one request and wait for its response."""
        with self._lock:
            request_id = self._next_id
            self._next_id += 1
            request = {"jsonrpc": "2.0", "method": method, "id": request_id}
            if params is not None:
                request["params"] = params
            payload = json.dumps(request).encode()
            from_context().debug("hipstore request", method=method)
            self._sock.sendall(payload)
            response = self._read_response()
        if response.get("id") not in (request_id, None):
            raise RpcError(ERROR_INTERNAL, "response id mismatch")
        if "error" in response:
            err = response["error"] or {}
            raise RpcError(int(err.get("code", ERROR_INTERNAL)),
                           str(err.get("message", "unknown error")))
        return response.get("result")

    def _read_response(self) -> dict:
        while True:
            try:
                text = self._buffer.decode("utf-8", errors="strict")
            except UnicodeDecodeError as exc:
                # recv() may split a multi-byte sequence at the TAIL;
                # invalid bytes anywhere else can never become valid
                # by reading more — fail instead of waiting forever.
                if exc.start < len(self._buffer) - 4:
                    raise RpcError(
                        ERROR_INTERNAL,
                        "invalid UTF-8 in daemon response") from None
                text = ""
            stripped = text.lstrip()
            offset = len(text) - len(stripped)
            if stripped:
                try:
                    value, end = self._decoder.raw_decode(stripped)
                    # Slice in BYTES: offset/end are character counts and
                    # differ from byte counts for non-ASCII payloads.
                    consumed = len(text[:offset + end].encode("utf-8"))
                    self._buffer = self._buffer[consumed:]
                    return value
                except json.JSONDecodeError:
                    pass  # incomplete; read more
            chunk = self._sock.recv(65536)
            if not chunk:
                raise ConnectionError("hipstored closed the connection")
            self._buffer += chunk
