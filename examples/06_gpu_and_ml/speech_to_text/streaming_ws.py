# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/speech_to_text/streaming_ws.py"]
# ---
# # Real-time transcription over a WebSocket
#
# The streaming-STT serving shape (kyutai/parakeet family,
# reference: speech-to-text/streaming_kyutai_stt.py:334-390): a browser (here,
# the self-test client) streams raw audio frames over ONE WebSocket; the
# server relays frames through a `modal.Queue` to the GPU transcriber and
# pushes incremental transcripts back over the same socket as they firm up.
#
# The self-test drives the WS end-to-end with synthetic audio and asserts a
# transcript for every flushed utterance — the reference's in-example
# self-test idiom (webrtc_yolo_test.py:27-33).

import modal_examples_amd as modal

app = modal.App("example-streaming-ws")

SR = 16000
FRAME = SR // 10  # 100 ms frames, the usual browser chunk


@app.cls(gpu="mi355x")
class StreamTranscriber:
    """GPU peer: incremental decode of utterances as they arrive."""

    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.whisper.model import WhisperConfig
        from modal_examples_amd.models.whisper.pipeline import WhisperPipeline

        gpu = torch.cuda.is_available()
        cfg = WhisperConfig.large_v3() if gpu else WhisperConfig.small_test()
        self.pipe = WhisperPipeline(cfg, device="cuda" if gpu else "cpu",
                                    dtype=torch.bfloat16 if gpu else torch.float32)

    @modal.method()
    def transcribe(self, samples) -> str:
        import torch

        seg = torch.as_tensor(samples, dtype=torch.float32)
        return self.pipe.transcribe_text([seg], max_tokens=8)[0]


@app.function()
@modal.asgi_app(label="stt-ws")
def web():
    import json

    import numpy as np
    from fastapi import FastAPI, WebSocket

    w = FastAPI()
    transcriber = StreamTranscriber()

    # vanilla-JS mic-streaming page (the kyutai frontend role)
    from pathlib import Path

    front = Path(__file__).parent / "stt_frontend"

    @w.get("/")
    async def index():
        from fastapi.responses import HTMLResponse

        return HTMLResponse((front / "index.html").read_text())

    @w.get("/app.js")
    async def appjs():
        from fastapi.responses import Response

        return Response((front / "app.js").read_text(),
                        media_type="text/javascript")

    @w.websocket("/stream")
    async def stream(ws: WebSocket):
        """Protocol: binary frames = float32 PCM @16 kHz; text "flush" ends an
        utterance; text "close" ends the session.  Server sends JSON
        {"utterance": i, "text": ...} per flushed utterance."""
        import asyncio

        await ws.accept()
        buf: list = []
        utt = 0
        while True:
            msg = await ws.receive()
            if msg.get("bytes") is not None:
                buf.append(np.frombuffer(msg["bytes"], dtype=np.float32))
                continue
            cmd = msg.get("text")
            if cmd == "flush" and buf:
                samples = np.concatenate(buf)
                buf = []
                text = await asyncio.to_thread(
                    transcriber.transcribe.remote, samples)
                await ws.send_text(json.dumps({"utterance": utt, "text": text}))
                utt += 1
            elif cmd == "close" or msg.get("type") == "websocket.disconnect":
                break
        if msg.get("type") != "websocket.disconnect":
            await ws.close()

    return w


@app.local_entrypoint()
def self_test():
    """Round-trip: stream 2 synthetic utterances, expect 2 transcripts."""
    import json
    import time

    import numpy as np
    from starlette.testclient import TestClient

    from modal_examples_amd.web.ingress import build_ingress_app

    root = build_ingress_app(app)
    rng = np.random.default_rng(0)
    with TestClient(root) as client:
        page = client.get("/stt-ws/")
        assert page.status_code == 200 and "Streaming transcription" in page.text
        with client.websocket_connect("/stt-ws/stream") as ws:
            for utt in range(2):
                t0 = time.monotonic()
                audio = (rng.standard_normal(SR) * 0.3).astype(np.float32)
                for i in range(0, len(audio), FRAME):
                    ws.send_bytes(audio[i:i + FRAME].tobytes())
                ws.send_text("flush")
                reply = json.loads(ws.receive_text())
                dt = time.monotonic() - t0
                assert reply["utterance"] == utt
                assert isinstance(reply["text"], str)
                print(f"utterance {utt}: {dt*1000:.0f} ms -> {reply['text']!r}")
            ws.send_text("close")
    print("streaming WS self-test ok")
