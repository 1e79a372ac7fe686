# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/udp_video_detect.py"]
# ---
# # Low-latency video inference over UDP (the misc/quic hole-punch role)
#
# The QUIC-YOLO shape: peers avoid a relay by exchanging their datagram
# endpoints through a RENDEZVOUS (there a STUN-style hole punch; here a
# `modal.Dict` both sides can reach), then stream video frames peer-to-peer
# over UDP — lossy, unordered, no head-of-line blocking — into a GPU
# detector that answers on the same socket.

import modal_examples_amd as modal

app = modal.App("example-udp-video")

rendezvous = modal.Dict.from_name("udp-rendezvous", create_if_missing=True)


@app.function(gpu="mi355x", timeout=180)
def detector_peer(session: str, max_frames: int = 32) -> int:
    """GPU peer: binds UDP, registers its endpoint, answers frame datagrams
    with detection scores until a FIN datagram."""
    import json
    import socket
    import struct

    import numpy as np
    import torch
    import torch.nn as nn

    torch.manual_seed(0)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    net = nn.Sequential(
        nn.Conv2d(3, 32, 3, stride=2, padding=1), nn.SiLU(),
        nn.Conv2d(32, 1, 1),
    ).to(device).eval()

    sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    sock.bind(("127.0.0.1", 0))
    sock.settimeout(20)
    rendezvous.put(session, {"addr": sock.getsockname()})  # the hole punch
    n = 0
    while n < max_frames:
        data, peer = sock.recvfrom(65536)
        if data == b"FIN":
            sock.sendto(b"FIN-ACK", peer)
            break
        seq = struct.unpack("<I", data[:4])[0]
        frame = np.frombuffer(data[4:], dtype=np.uint8).astype("float32")
        side = int((len(frame) / 3) ** 0.5)
        x = torch.as_tensor(frame.reshape(3, side, side) / 255.0,
                            device=device)[None]
        with torch.no_grad():
            score = float(net(x).max())
        sock.sendto(json.dumps({"seq": seq, "score": round(score, 3)}).encode(),
                    peer)
        n += 1
    return n


@app.local_entrypoint()
def main(frames: int = 12):
    import json
    import socket
    import struct
    import time

    import numpy as np

    import uuid

    session = f"sess-{uuid.uuid4().hex[:8]}"  # unique per run: a stale peer
    # from an interrupted previous run must not re-register our key
    rendezvous.delete(session)
    call = detector_peer.spawn(session)
    # rendezvous: wait for the peer's punched endpoint
    deadline = time.time() + 30
    info = None
    while time.time() < deadline and info is None:
        info = rendezvous.get(session)
        time.sleep(0.1)
    assert info, "peer never registered"
    addr = tuple(info["addr"])

    sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    sock.settimeout(10)
    rng = np.random.default_rng(0)
    lat, got = [], 0
    for seq in range(frames):
        frame = rng.integers(0, 255, 3 * 32 * 32, dtype=np.uint8)
        t0 = time.monotonic()
        sock.sendto(struct.pack("<I", seq) + frame.tobytes(), addr)
        try:
            reply = json.loads(sock.recv(65536))
            got += 1
            lat.append((time.monotonic() - t0) * 1000)
            assert reply["seq"] == seq
        except socket.timeout:
            pass  # UDP: losses are part of the contract
    sock.sendto(b"FIN", addr)
    served = call.get(timeout=30)
    assert got >= frames - 2, (got, frames)  # tolerance, reference idiom
    print(f"{got}/{frames} frames over UDP, median rtt "
          f"{sorted(lat)[len(lat)//2]:.1f} ms; peer served {served}")
