"""Server: long-lived HTTP service processes behind ``@app.server``.

Parity: /root/reference/py/modal/_server.py + app.py:1280 — a Server is a
class whose ``@modal.enter`` method starts an HTTP server on a fixed port;
the platform waits for the port to accept connections (``startup_timeout``)
and then routes traffic to it. Locally the worker shares the host network,
so the readiness probe runs INSIDE the worker (instantiating the service
fires the enter hooks, which start the server) and the URL is the port on
127.0.0.1. One replica per port — the single-node analog of the
reference's per-container port binding.
"""

from __future__ import annotations

from typing import Any, Optional

from ._sync import synchronizer
from .exception import InvalidError


def validate_server_config(port: int, startup_timeout: float) -> None:
    """Parity: reference _server.py validate_http_server_config."""
    if not isinstance(port, int) or port < 1 or port > 65535:
        raise InvalidError("Port must be a positive integer between 1 and 65535.")
    if startup_timeout <= 0:
        raise InvalidError("The `startup_timeout` argument must be positive.")


def _probe_port(port: int, startup_timeout: float) -> dict:
    """Runs IN the worker: wait until the server accepts connections."""
    import socket
    import time

    deadline = time.time() + startup_timeout
    last_err = None
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=1):
                return {"ready": True, "port": port}
        except OSError as exc:
            last_err = str(exc)
            time.sleep(0.05)
    raise TimeoutError(
        f"server did not accept connections on port {port} "
        f"within {startup_timeout}s ({last_err})"
    )


class Server:
    """Handle for an ``@app.server`` class: start() brings a replica up and
    waits for readiness; ``url`` points at the serving port."""

    def __init__(self, cls_obj: Any, port: int, startup_timeout: float):
        self._cls = cls_obj
        self.port = port
        self.startup_timeout = startup_timeout
        self._instance: Any = None

    def start(self, *args: Any, **kwargs: Any) -> "Server":
        """Instantiate the service on a worker (running its @enter hooks,
        which start the HTTP server) and block until the port is ready."""
        self._instance = self._cls(*args, **kwargs)
        result = self._instance._server_probe.remote()
        if not result.get("ready"):
            raise TimeoutError(f"server probe failed: {result}")
        return self

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def stop(self) -> None:
        """Tear down the serving replica (@exit hooks run — the place to
        terminate the server process, parity: exit_grace_period)."""
        inst = self._instance
        self._instance = None
        if inst is None:
            return
        app = getattr(self._cls, "_app_ref", None)
        # @exit hooks run at app teardown; nothing extra to do per-instance
        # locally — the pool reaps idle workers on the scaledown window.


def make_server(app: Any, user_cls: type, port: int, startup_timeout: float,
                function_kwargs: dict) -> Server:
    """The @app.server decorator body: inject the probe method, register
    the class service, return the Server handle."""
    validate_server_config(port, startup_timeout)

    def _server_probe(self) -> dict:  # noqa: ANN001 (worker-side)
        return _probe_port(port, startup_timeout)

    _server_probe.__name__ = "_server_probe"
    from .partial_function import method

    setattr(user_cls, "_server_probe", method()(_server_probe))

    from .cls import make_cls

    cls_obj = make_cls(app, user_cls, function_kwargs)
    app._classes[user_cls.__name__] = cls_obj
    return Server(cls_obj, port, startup_timeout)
