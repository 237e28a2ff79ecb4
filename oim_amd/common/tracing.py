"""gRPC payload-logging interceptors (reference pkg/oim-common/tracing.go).

Server- and client-side unary interceptors that log method, payload and
outcome through oim_amd.log, with pluggable payload formatters:
``complete`` (full payload), ``strip_secrets`` (redacts fields named in
SECRET_FIELDS — the protosanitizer analog) and ``null`` (no payloads).
"""

from __future__ import annotations

from typing import Any, Callable

import grpc

from ..log import from_context

SECRET_FIELDS = {"secrets", "user_key", "admin_key", "key"}


def complete_formatter(message: Any) -> str:
    return _render(message, strip=False)


def strip_secrets_formatter(message: Any) -> str:
    return _render(message, strip=True)


def null_formatter(message: Any) -> str:
    return "<omitted>"


def _render(message: Any, strip: bool) -> str:
    try:
        fields = []
        for descriptor, value in message.ListFields():
            if strip and descriptor.name in SECRET_FIELDS:
                fields.append(f"{descriptor.name}=<redacted>")
            else:
                fields.append(f"{descriptor.name}={value!r}")
        return "{" + ", ".join(fields) + "}"
    except AttributeError:
        return repr(message)


class LogServerInterceptor(grpc.ServerInterceptor):
    """Logs every incoming call (tracing.go:92-110)."""

    def __init__(self, formatter: Callable[[Any], str] = complete_formatter):
        self.formatter = formatter

    def intercept_service(self, continuation, handler_call_details):
        handler = continuation(handler_call_details)
        if handler is None or not handler.unary_unary:
            return handler
        method = handler_call_details.method
        formatter = self.formatter
        inner = handler.unary_unary

        def wrapper(request, context):
            log = from_context()
            log.debug("request", method=method, payload=formatter(request))
            try:
                response = inner(request, context)
            except Exception as exc:  # noqa: BLE001 - log and re-raise
                log.error("request failed", method=method, error=str(exc))
                raise
            log.debug("response", method=method, payload=formatter(response))
            return response

        return grpc.unary_unary_rpc_method_handler(
            wrapper,
            request_deserializer=handler.request_deserializer,
            response_serializer=handler.response_serializer,
        )


class LogClientInterceptor(grpc.UnaryUnaryClientInterceptor):
    """Logs every outgoing call (tracing.go:114-132).

    Defaults to secret stripping like the reference's client dial options
    (grpc.go:56).
    """

    def __init__(self, formatter: Callable[[Any], str] = strip_secrets_formatter):
        self.formatter = formatter

    def intercept_unary_unary(self, continuation, client_call_details, request):
        log = from_context()
        log.debug(
            "invoking",
            method=client_call_details.method,
            payload=self.formatter(request),
        )
        return continuation(client_call_details, request)
