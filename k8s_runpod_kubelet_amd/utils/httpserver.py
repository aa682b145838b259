"""ThreadingHTTPServer that logs client disconnects instead of dumping
socketserver tracebacks to stderr (kubectl/httpx hanging up mid-request —
Ctrl-C on `kubectl logs -f`, pool recycling — is normal operation, not an
error worth a stack trace)."""

from __future__ import annotations

import logging
from http.server import ThreadingHTTPServer

log = logging.getLogger("utils.httpserver")


class QuietThreadingHTTPServer(ThreadingHTTPServer):
    daemon_threads = True

    def handle_error(self, request, client_address):
        import sys

        exc = sys.exc_info()[1]
        if isinstance(exc, (ConnectionResetError, BrokenPipeError,
                            ConnectionAbortedError, TimeoutError)):
            log.debug("client disconnected", extra={
                "peer": str(client_address), "err": str(exc)})
            return
        super().handle_error(request, client_address)
