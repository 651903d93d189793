"""@app.server: a long-lived HTTP service behind the class-service
machinery — @enter starts the server process, start() waits for the port
(parity: the reference's Server / guide/servers).

Run:  modal-amd run examples/http_server_class.py::app.main
"""

import modal_amd as modal

app = modal.App("example-server")


@app.server(port=8199, startup_timeout=20)
class EchoServer:
    @modal.enter()
    def boot(self):
        import subprocess
        import sys

        self.proc = subprocess.Popen(
            [
                sys.executable, "-c",
                "import http.server\n"
                "class H(http.server.BaseHTTPRequestHandler):\n"
                "    def do_GET(self):\n"
                "        self.send_response(200); self.end_headers()\n"
                "        self.wfile.write(b'served by EchoServer')\n"
                "    def log_message(self, *a): pass\n"
                "http.server.HTTPServer(('127.0.0.1', 8199), H).serve_forever()",
            ]
        )

    @modal.exit()
    def shutdown(self):
        self.proc.terminate()


@app.local_entrypoint()
def main():
    import urllib.request

    server = EchoServer.start()
    print("serving at", server.url)
    with urllib.request.urlopen(server.url, timeout=10) as resp:
        print("response:", resp.read().decode())
    # expose it through a tunnel relay as well
    with modal.forward(8199, unencrypted=True) as tunnel:
        host, port = tunnel.tcp_socket
        with urllib.request.urlopen(f"http://{host}:{port}", timeout=10) as resp:
            print("via tunnel:", resp.read().decode())
    server.stop()
