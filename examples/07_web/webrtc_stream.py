# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/webrtc_stream.py"]
# ---
# # Real-time frame streaming with Queue signaling
#
# The WebRTC-YOLO shape without the browser: a signaling channel over
# `modal.Queue` negotiates a session between a "peer" (frame source) and a
# GPU processor; frames stream through a partitioned queue and processed
# detections stream back, with per-frame latency measured.

import time

import modal_examples_amd as modal

app = modal.App("example-webrtc-stream")

signaling = modal.Queue.from_name("rtc-signaling", create_if_missing=True)
media = modal.Queue.from_name("rtc-media", create_if_missing=True)


@app.cls(gpu="mi355x")
@modal.concurrent(max_inputs=4)
class FrameProcessor:
    @modal.enter()
    def load(self):
        import torch
        import torch.nn as nn

        self.torch = torch
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        torch.manual_seed(0)
        # detector stand-in: conv backbone → grid of objectness scores
        self.net = nn.Sequential(
            nn.Conv2d(3, 32, 3, stride=2, padding=1), nn.SiLU(),
            nn.Conv2d(32, 64, 3, stride=2, padding=1), nn.SiLU(),
            nn.Conv2d(64, 1, 1),
        ).to(self.device)

    @modal.method()
    def serve_session(self, session_id: str, max_frames: int = 16) -> dict:
        """Answer one peer: read offer, process frames until 'bye'."""
        from queue import Empty

        torch = self.torch
        try:
            offer = signaling.get(partition=session_id, timeout=30)
        except Empty:
            return {"frames": 0, "error": "no offer"}
        signaling.put({"type": "answer", "codec": offer["codec"]},
                      partition=f"{session_id}-answer")
        n, lat = 0, []
        while n < max_frames:
            try:
                msg = media.get(partition=session_id, timeout=10)
            except Empty:
                break
            if msg is None or msg.get("type") == "bye":
                break
            x = torch.as_tensor(msg["frame"], dtype=torch.float32,
                                device=self.device).unsqueeze(0)
            with torch.no_grad():
                scores = self.net(x)[0, 0]
            k = min(3, scores.numel())
            top = torch.topk(scores.flatten(), k)
            dets = [{"score": round(float(s), 3), "cell": int(i)}
                    for s, i in zip(top.values, top.indices)]
            media.put({"frame_id": msg["frame_id"], "detections": dets,
                       "t_sent": msg["t_sent"]},
                      partition=f"{session_id}-out")
            n += 1
        return {"frames": n}


@app.local_entrypoint()
def main(frames: int = 6):
    import numpy as np

    session = "sess-demo"
    signaling.clear(all=True)
    media.clear(all=True)
    proc = FrameProcessor()
    handle = proc.serve_session.spawn(session, frames)

    # peer side: offer → answer → stream frames
    signaling.put({"type": "offer", "codec": "raw-rgb"}, partition=session)
    from queue import Empty

    ans = None
    for _ in range(12):  # worker cold start may take a few seconds
        try:
            ans = signaling.get(partition=f"{session}-answer", timeout=10)
            break
        except Empty:
            continue
    assert ans is not None, "no answer from processor"
    print("negotiated:", ans)
    rng = np.random.default_rng(0)
    for i in range(frames):
        media.put({"type": "frame", "frame_id": i, "t_sent": time.time(),
                   "frame": rng.standard_normal((3, 64, 64)).astype("float32")},
                  partition=session)
    media.put({"type": "bye"}, partition=session)
    got = 0
    while got < frames:
        try:
            out = media.get(partition=f"{session}-out", timeout=30)
        except Empty:
            break
        if out is None:
            break
        rtt = (time.time() - out["t_sent"]) * 1000
        print(f"frame {out['frame_id']}: {len(out['detections'])} detections, "
              f"rtt {rtt:.0f} ms")
        got += 1
    print("processor:", handle.get())
    assert got == frames
