# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/flask_app.py"]
# ---
# # Serving a Flask app
#
# The Flask shape (reference: 07_web/flask_app.py + flask_streaming.py):
# a stock Flask application mounted through `@modal.wsgi_app` — the WSGI
# adapter wraps it for the shared uvicorn ingress, including a streaming
# route.  The entrypoint self-tests both routes over an in-process client.

import modal_examples_amd as modal

app = modal.App("example-flask")


@app.function()
@modal.wsgi_app(label="flask")
def flask_factory():
    from flask import Flask, Response, request

    web = Flask("mi355x-flask")

    @web.get("/")
    def home():
        return {"framework": "flask", "runtime": "modal_examples_amd"}

    @web.post("/echo")
    def echo():
        data = request.get_json(force=True, silent=True) or {}
        return {"echo": data, "args": dict(request.args)}

    @web.get("/stream")
    def stream():
        def gen():
            for i in range(5):
                yield f"chunk-{i}\n"

        return Response(gen(), mimetype="text/plain")

    return web


@app.local_entrypoint()
def main():
    wsgi = flask_factory.raw()
    from werkzeug.test import Client

    c = Client(wsgi)
    r = c.get("/")
    assert r.status_code == 200 and r.get_json()["framework"] == "flask"
    r = c.post("/echo?tag=x", json={"a": 1})
    body = r.get_json()
    assert body["echo"] == {"a": 1} and body["args"] == {"tag": "x"}
    r = c.get("/stream")
    chunks = r.get_data(as_text=True).splitlines()
    assert chunks == [f"chunk-{i}" for i in range(5)]
    print("flask routes:", body, "| stream:", chunks)
    print("flask-over-wsgi OK")
