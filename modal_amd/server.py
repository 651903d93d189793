"""Server: long-lived service processes behind @app.server.

Parity: /root/reference/py/modal/server.py + app.py:1280 — a Server is a
class service whose instance keeps serving (web or custom protocol); locally
it maps onto the class-service machinery with @enter-started serving.
"""

from __future__ import annotations

from .cls import Cls


class Server(Cls):
    """Alias of the class-service handle with server semantics."""
