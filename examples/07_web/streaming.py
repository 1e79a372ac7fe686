# ---
# cmd: ["python", "-m", "modal_examples_amd", "serve", "examples/07_web/streaming.py", "--timeout", "3"]
# ---
# # Server-sent events
#
# Stream responses incrementally: an ASGI app yields SSE chunks fed by a
# remote generator function (`.remote_gen` inside the response body).

import time

import modal_examples_amd as modal

app = modal.App("example-streaming")


@app.function()
def compute_chunks(n: int):
    for i in range(n):
        time.sleep(0.05)
        yield f"chunk {i}"


@app.function()
@modal.asgi_app(label="sse")
def sse_app():
    from fastapi import FastAPI
    from fastapi.responses import StreamingResponse

    web = FastAPI()

    @web.get("/stream")
    def stream(n: int = 5):
        def gen():
            for item in compute_chunks.remote_gen(n):
                yield f"data: {item}\n\n"
            yield "data: [DONE]\n\n"

        return StreamingResponse(gen(), media_type="text/event-stream")

    return web


@app.local_entrypoint()
def main():
    import httpx

    from modal_examples_amd.web.ingress import serve, stop_serving

    base = serve(app, port=8795)
    try:
        with httpx.stream("GET", f"{base}/sse/stream", params={"n": 3},
                          timeout=60) as r:
            lines = [l for l in r.iter_lines() if l.startswith("data:")]
        print("\n".join(lines))
        assert lines[-1] == "data: [DONE]" and len(lines) == 4
    finally:
        stop_serving()
