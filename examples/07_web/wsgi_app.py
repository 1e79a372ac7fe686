# ---
# cmd: ["python", "-m", "modal_examples_amd", "serve", "examples/07_web/wsgi_app.py", "--timeout", "3"]
# ---
# # WSGI apps
#
# `@modal.wsgi_app` mounts any WSGI callable on the ingress (the flask
# pattern; shown here dependency-free with a plain WSGI function).

import json

import modal_examples_amd as modal

app = modal.App("example-wsgi")


@app.function()
@modal.wsgi_app(label="legacy")
def legacy_app():
    def application(environ, start_response):
        path = environ.get("PATH_INFO", "/")
        body = json.dumps({
            "path": path,
            "method": environ.get("REQUEST_METHOD"),
            "message": "served by a WSGI app on the MI355X runner",
        }).encode()
        start_response("200 OK", [("Content-Type", "application/json"),
                                  ("Content-Length", str(len(body)))])
        return [body]

    return application


@app.local_entrypoint()
def main():
    import httpx

    from modal_examples_amd.web.ingress import serve, stop_serving

    base = serve(app, port=8796)
    try:
        r = httpx.get(f"{base}/legacy/hello", timeout=30)
        print(r.json())
        assert r.json()["path"] == "/hello"
    finally:
        stop_serving()
