# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/13_sandboxes/sandbox_web_service.py"]
# ---
# # A web service inside a Sandbox, reached through its tunnel
#
# The opencode/jupyter-in-sandbox shape (13_sandboxes/opencode_server.py,
# jupyter_sandbox.py): start a long-running server INSIDE a sandbox on an
# encrypted port, wait for readiness, talk to it through `sb.tunnels()`,
# then detach/terminate.  The server here is a small JSON API the sandbox
# hosts from its own scratch filesystem.

import modal_examples_amd as modal

app = modal.App("example-sandbox-web")

SERVER = r"""
import http.server, json, os

class H(http.server.BaseHTTPRequestHandler):
    def do_GET(self):
        body = json.dumps({
            "path": self.path,
            "cwd": os.getcwd(),
            "files": sorted(os.listdir(".")),
        }).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.end_headers()
        self.wfile.write(body)

    def log_message(self, *a):
        pass

http.server.HTTPServer(("127.0.0.1", PORT), H).serve_forever()
"""


@app.local_entrypoint()
def main():
    import json
    import socket
    import urllib.request

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()

    sb = modal.Sandbox.create(
        "python", "-c", SERVER.replace("PORT", str(port)),
        timeout=120, encrypted_ports=[port],
        readiness_probe=modal.Probe.with_exec(
            ["python", "-c",
             f"import urllib.request;urllib.request.urlopen('http://127.0.0.1:{port}/')"]))
    try:
        assert sb.wait_until_ready(30)
        with sb.open("hello.txt", "w") as f:  # file lands in the sandbox fs
            f.write("from the host")
        tunnel = sb.tunnels()[port]
        with urllib.request.urlopen(
                f"http://{tunnel.host}:{tunnel.port}/status", timeout=5) as r:
            info = json.loads(r.read())
        assert info["path"] == "/status"
        assert "hello.txt" in info["files"], info
        print(f"sandbox service via tunnel {tunnel.host}:{tunnel.port}: {info}")
    finally:
        sb.terminate()
