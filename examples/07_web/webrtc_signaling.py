# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/webrtc_signaling.py"]
# ---
# # WebRTC-style signaling: WebSocket server relaying SDP to a GPU peer
#
# The reference's `ModalWebRtcSignalingServer` / `ModalWebRtcPeer` split
# (07_web/webrtc/modal_webrtc.py:16-100): a browser opens a WEBSOCKET to the
# signaling server, which relays the offer/answer/ICE negotiation over a
# `modal.Queue` to a GPU peer container spawned per session; once negotiated,
# media flows peer-to-peer (here: frames over the established channel, with a
# YOLO-role conv detector on the GPU side, webrtc_yolo.py:93 2-4 ms/frame).
#
# The self-test plays the browser: WS connect → send offer → receive answer
# (negotiated by the GPU peer through the queue) → stream frames → assert
# detections + frame-count tolerance (webrtc_yolo_test.py:27-33 idiom).

import modal_examples_amd as modal

app = modal.App("example-webrtc-signaling")

signaling = modal.Queue.from_name("rtc-ws-signaling", create_if_missing=True)


@app.cls(gpu="mi355x", timeout=300)
class GpuPeer:
    """The ModalWebRtcPeer role: answers SDP offers, then processes media."""

    @modal.enter()
    def load(self):
        import torch
        import torch.nn as nn

        self.torch = torch
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        torch.manual_seed(0)
        self.net = nn.Sequential(
            nn.Conv2d(3, 32, 3, stride=2, padding=1), nn.SiLU(),
            nn.Conv2d(32, 64, 3, stride=2, padding=1), nn.SiLU(),
            nn.Conv2d(64, 1, 1),
        ).to(self.device).eval()

    @modal.method()
    def run_session(self, session_id: str, max_frames: int = 64) -> int:
        """Negotiate (answer the offer), then detect on incoming frames."""
        import numpy as np
        import torch

        # --- signaling: wait for the relayed offer, post the answer
        offer = signaling.get(partition=f"{session_id}:to-peer")
        assert offer["type"] == "offer", offer
        answer = {"type": "answer", "sdp": f"v=0 answer-for {offer['sdp'][:24]}",
                  "session": session_id}
        signaling.put(answer, partition=f"{session_id}:to-client")
        # --- ICE candidate exchange (one mock candidate each way)
        cand = signaling.get(partition=f"{session_id}:to-peer")
        assert cand["type"] == "ice"
        signaling.put({"type": "ice", "candidate": "peer-host"},
                      partition=f"{session_id}:to-client")

        # --- media: frames arrive on the negotiated channel
        n = 0
        while n < max_frames:
            import queue as _q

            try:  # Queue.get(timeout=) raises Empty (reference semantics)
                msg = signaling.get(partition=f"{session_id}:media-in",
                                    timeout=10)
            except _q.Empty:
                break
            if msg.get("type") == "bye":
                break
            frame = torch.as_tensor(np.asarray(msg["frame"], dtype="float32"))
            with torch.no_grad():
                score = self.net(frame[None].to(self.device))
            det = float(score.max())
            signaling.put({"type": "detection", "frame": msg["seq"],
                           "score": round(det, 3)},
                          partition=f"{session_id}:media-out")
            n += 1
        return n


@app.function()
@modal.asgi_app(label="rtc")
def signaling_server():
    """The ModalWebRtcSignalingServer role: one WS per client session; spawns
    the GPU peer and relays both directions through the Queue."""
    import asyncio
    import json
    import uuid

    from fastapi import FastAPI, WebSocket

    w = FastAPI()
    peer_cls = GpuPeer()

    @w.websocket("/ws")
    async def ws_session(ws: WebSocket):
        await ws.accept()
        session_id = uuid.uuid4().hex[:8]
        call = peer_cls.run_session.spawn(session_id)  # one peer per session
        await ws.send_text(json.dumps({"type": "session", "id": session_id}))
        try:
            while True:
                raw = await ws.receive_text()
                msg = json.loads(raw)
                if msg.get("type") in ("offer", "ice"):
                    signaling.put(msg, partition=f"{session_id}:to-peer")
                    reply = await asyncio.to_thread(
                        signaling.get, partition=f"{session_id}:to-client",
                        timeout=30)
                    await ws.send_text(json.dumps(reply))
                elif msg.get("type") == "frame":
                    signaling.put(msg, partition=f"{session_id}:media-in")
                    det = await asyncio.to_thread(
                        signaling.get, partition=f"{session_id}:media-out",
                        timeout=30)
                    await ws.send_text(json.dumps(det))
                elif msg.get("type") == "bye":
                    signaling.put(msg, partition=f"{session_id}:media-in")
                    break
        finally:
            try:
                call.get(timeout=30)
            except Exception:
                pass
        await ws.close()

    return w


@app.local_entrypoint()
def main(frames: int = 6):
    import json
    import time

    import numpy as np
    from starlette.testclient import TestClient

    from modal_examples_amd.web.ingress import build_ingress_app

    rng = np.random.default_rng(0)
    root = build_ingress_app(app)
    with TestClient(root) as client:
        with client.websocket_connect("/rtc/ws") as ws:
            sess = json.loads(ws.receive_text())
            assert sess["type"] == "session"
            ws.send_text(json.dumps({"type": "offer", "sdp": "v=0 client-offer"}))
            answer = json.loads(ws.receive_text())
            assert answer["type"] == "answer" and "answer-for" in answer["sdp"]
            ws.send_text(json.dumps({"type": "ice", "candidate": "client-host"}))
            ice = json.loads(ws.receive_text())
            assert ice["type"] == "ice"
            print(f"negotiated session {sess['id']}: {answer['sdp']!r}")

            lat = []
            got = 0
            for seq in range(frames):
                frame = rng.standard_normal((3, 64, 64)).astype("float32")
                t0 = time.monotonic()
                ws.send_text(json.dumps({"type": "frame", "seq": seq,
                                         "frame": frame.tolist()}))
                det = json.loads(ws.receive_text())
                lat.append((time.monotonic() - t0) * 1000)
                if det.get("type") == "detection":
                    got += 1
            ws.send_text(json.dumps({"type": "bye"}))
    # frame-count tolerance, the reference self-test idiom
    assert got >= frames - 1, (got, frames)
    print(f"{got}/{frames} frames detected; median rtt "
          f"{sorted(lat)[len(lat)//2]:.0f} ms")
