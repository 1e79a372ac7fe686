// Minimal dependency-free client for the receipt-parser job queue:
// POST /api/submit -> {call_id}, then poll GET /api/status?call_id=... until
// the worker finishes (the doc-OCR webapp flow, without a build toolchain).

const $ = (sel) => document.querySelector(sel);

const POLL_MS = 500;

async function submitJob(text) {
  const r = await fetch("api/submit", {
    method: "POST",
    headers: { "Content-Type": "application/json" },
    body: JSON.stringify({ text }),
  });
  if (!r.ok) throw new Error(`submit failed: ${r.status}`);
  return (await r.json()).call_id;
}

async function pollJob(callId) {
  const r = await fetch(`api/status?call_id=${encodeURIComponent(callId)}`);
  if (!r.ok) throw new Error(`status failed: ${r.status}`);
  return r.json();
}

function renderResult(parsed) {
  const rows = (parsed.items || [])
    .map((it) => `<tr><td>${it.qty}×</td><td>${it.name}</td><td>$${it.price.toFixed(2)}</td></tr>`)
    .join("");
  return `<table>${rows}</table><div>total: <b>$${(parsed.total ?? 0).toFixed(2)}</b>
    (${parsed.n_lines} lines scanned)</div>`;
}

function jobCard(callId) {
  const el = document.createElement("div");
  el.className = "job";
  el.innerHTML = `<div>job <code>${callId}</code>
    <span class="status pending">pending…</span></div><div class="body"></div>`;
  $("#jobs").prepend(el);
  return el;
}

async function runJob(text) {
  const btn = $("#submit");
  btn.disabled = true;
  try {
    const callId = await submitJob(text);
    const card = jobCard(callId);
    const status = card.querySelector(".status");
    for (;;) {
      const s = await pollJob(callId);
      if (s.status === "done") {
        status.textContent = "done";
        status.className = "status done";
        card.querySelector(".body").innerHTML = renderResult(s.result);
        break;
      }
      if (s.status === "error") {
        status.textContent = `error: ${s.detail}`;
        status.className = "status error";
        break;
      }
      await new Promise((res) => setTimeout(res, POLL_MS));
    }
  } finally {
    btn.disabled = false;
  }
}

$("#submit").addEventListener("click", () => {
  const text = $("#doc").value.trim();
  if (text) runJob(text);
});
