# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/raw_port_server.py"]
# ---
# # Raw-port servers and the 503-until-warm client
#
# `@modal.web_server(port)` runs a server the function binds itself; clients
# retry until the container is warm (the Modal Server pattern).

import time

import modal_examples_amd as modal

app = modal.App("example-raw-server")

PORT = 8973


@app.cls(scaledown_window=60)
class EchoServer:
    @modal.enter()
    def boot(self):
        import http.server
        import threading

        class Handler(http.server.BaseHTTPRequestHandler):
            def do_GET(self):
                time.sleep(0.01)
                body = f"echo:{self.path}".encode()
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):
                pass

        self.httpd = http.server.ThreadingHTTPServer(("127.0.0.1", PORT), Handler)
        threading.Thread(target=self.httpd.serve_forever, daemon=True).start()

    @modal.web_server(port=PORT, startup_timeout=30)
    def serve(self):
        pass  # server already bound in @enter

    @modal.method()
    def ping(self) -> str:
        import urllib.request

        return urllib.request.urlopen(
            f"http://127.0.0.1:{PORT}/warm-check", timeout=5).read().decode()

    @modal.exit()
    def stop(self):
        self.httpd.shutdown()


@app.local_entrypoint()
def main():
    server = EchoServer()
    # 503-until-warm loop: retry until the container answers
    for attempt in range(30):
        try:
            out = server.ping.remote()
            break
        except Exception:
            time.sleep(0.5)
    else:
        raise SystemExit("server never warmed")
    print(f"warm after {attempt + 1} attempt(s): {out}")
    assert out == "echo:/warm-check"
