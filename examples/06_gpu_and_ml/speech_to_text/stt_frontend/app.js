// Browser side of the streaming-STT protocol (streaming_ws.py):
// binary frames = float32 PCM @16kHz, "flush" ends an utterance, server
// replies {"utterance": i, "text": ...} per flush.
const logEl = document.getElementById("log");
const startBtn = document.getElementById("start");
const stopBtn = document.getElementById("stop");
let ws, ctx, node, silentFrames = 0, streaming = false;

function log(html) {
  const d = document.createElement("div");
  d.className = "utt";
  d.innerHTML = html;
  logEl.appendChild(d);
}

startBtn.onclick = async () => {
  const proto = location.protocol === "https:" ? "wss" : "ws";
  ws = new WebSocket(`${proto}://${location.host}/stt-ws/stream`);
  ws.onmessage = (ev) => {
    const msg = JSON.parse(ev.data);
    log(`<b>#${msg.utterance}</b> ${msg.text || "<i>(silence)</i>"}`);
  };
  const stream = await navigator.mediaDevices.getUserMedia({ audio: true });
  ctx = new AudioContext({ sampleRate: 16000 });
  const src = ctx.createMediaStreamSource(stream);
  node = ctx.createScriptProcessor(1600, 1, 1); // 100 ms frames
  node.onaudioprocess = (e) => {
    if (!streaming || ws.readyState !== 1) return;
    const pcm = e.inputBuffer.getChannelData(0);
    ws.send(new Float32Array(pcm).buffer);
    // crude endpointing: flush after ~600 ms of low energy
    const rms = Math.sqrt(pcm.reduce((s, v) => s + v * v, 0) / pcm.length);
    silentFrames = rms < 0.01 ? silentFrames + 1 : 0;
    if (silentFrames === 6) ws.send("flush");
  };
  src.connect(node);
  node.connect(ctx.destination);
  streaming = true;
  startBtn.disabled = true;
  stopBtn.disabled = false;
  log('<span class="meta">mic open — speak, pause to flush an utterance</span>');
};

stopBtn.onclick = () => {
  streaming = false;
  if (ws && ws.readyState === 1) { ws.send("flush"); ws.send("close"); }
  if (ctx) ctx.close();
  startBtn.disabled = false;
  stopBtn.disabled = true;
};
