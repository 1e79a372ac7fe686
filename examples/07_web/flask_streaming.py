# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/flask_streaming.py"]
# ---
# # Streaming responses from Flask (07_web/flask_streaming.py role)
#
# A Flask wsgi app whose response body streams from a `.remote_gen` generator
# running in a GPU-capable worker — chunked transfer out of a wsgi app.

import modal_examples_amd as modal

app = modal.App("example-flask-streaming")


@app.function()
def progress(n: int = 5):
    import time

    for i in range(n):
        time.sleep(0.02)
        yield f"step {i}\n"


@app.function()
@modal.wsgi_app(label="flaskstream")
def web():
    from flask import Flask, Response

    f = Flask("stream")

    @f.get("/run")
    def run():
        return Response((chunk for chunk in progress.remote_gen(5)),
                        mimetype="text/plain")

    return f


@app.local_entrypoint()
def main():
    import asyncio

    import httpx

    from modal_examples_amd.web.ingress import build_ingress_app

    async def go():
        root = build_ingress_app(app)
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            r = await c.get("/flaskstream/run")
            return r.text

    body = asyncio.run(go())
    assert body.splitlines() == [f"step {i}" for i in range(5)]
    print("flask streamed 5 chunks from a remote generator")
