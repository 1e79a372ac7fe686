// Minimal OpenAI-compatible chat client: POST /v1/chat/completions with
// stream=true and render SSE deltas token-by-token (the llm-frontend role,
// as plain ES modules — no build step).

const log = document.querySelector("#log");
const form = document.querySelector("#form");
const box = document.querySelector("#box");
const send = document.querySelector("#send");

const history = [];

fetch("v1/models").then((r) => r.json()).then((d) => {
  document.querySelector("#model").textContent = d.data?.[0]?.id ?? "";
});

function bubble(role, text = "") {
  const el = document.createElement("div");
  el.className = `msg ${role}`;
  el.textContent = text;
  log.appendChild(el);
  log.scrollTop = log.scrollHeight;
  return el;
}

async function chat(prompt) {
  history.push({ role: "user", content: prompt });
  bubble("user", `you: ${prompt}`);
  const el = bubble("assistant", "model: ");
  const t0 = performance.now();
  let ntok = 0;

  const resp = await fetch("v1/chat/completions", {
    method: "POST",
    headers: { "Content-Type": "application/json" },
    body: JSON.stringify({ messages: history, stream: true, max_tokens: 64 }),
  });
  const reader = resp.body.getReader();
  const dec = new TextDecoder();
  let buf = "", text = "";
  for (;;) {
    const { value, done } = await reader.read();
    if (done) break;
    buf += dec.decode(value, { stream: true });
    let idx;
    while ((idx = buf.indexOf("\n\n")) >= 0) {
      const frame = buf.slice(0, idx).trim();
      buf = buf.slice(idx + 2);
      if (!frame.startsWith("data:")) continue;
      const data = frame.slice(5).trim();
      if (data === "[DONE]") continue;
      const delta = JSON.parse(data).choices?.[0]?.delta?.content ?? "";
      text += delta;
      ntok += 1;
      el.textContent = `model: ${text}`;
      log.scrollTop = log.scrollHeight;
    }
  }
  history.push({ role: "assistant", content: text });
  const dt = (performance.now() - t0) / 1000;
  const meta = document.createElement("span");
  meta.className = "meta";
  meta.textContent = `  ${ntok} tok · ${(ntok / dt).toFixed(1)} tok/s`;
  el.appendChild(meta);
}

form.addEventListener("submit", async (e) => {
  e.preventDefault();
  const prompt = box.value.trim();
  if (!prompt) return;
  box.value = "";
  send.disabled = true;
  try {
    await chat(prompt);
  } finally {
    send.disabled = false;
    box.focus();
  }
});
