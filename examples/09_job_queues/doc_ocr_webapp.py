# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/09_job_queues/doc_ocr_webapp.py"]
# ---
# # Document-parser webapp: JS frontend + job queue backend
# (reference: 09_job_queues/doc_ocr_webapp.py + doc_ocr_frontend/app.jsx)
#
# The doc-OCR webapp shape: a static JS single-page app (doc_ocr_frontend/)
# submits documents to a REST API that spawns parse jobs and polls results.
# The "OCR" is a deterministic receipt parser so the example is hermetic.
# The entrypoint self-tests the whole stack (HTML, JS asset, submit → poll →
# parsed result) over an in-process ASGI transport — no browser needed.

import re
from pathlib import Path

import modal_examples_amd as modal

app = modal.App("example-doc-ocr-webapp")

FRONTEND = Path(__file__).parent / "doc_ocr_frontend"

LINE = re.compile(r"^\s*(?:(\d+)\s*x?\s+)?(.*?)\s+\$?(\d+(?:\.\d{1,2})?)\s*$",
                  re.IGNORECASE)


@app.function()
def parse_receipt(text: str) -> dict:
    """Worker job: extract line items and the total from receipt text."""
    items, total = [], None
    lines = [ln for ln in text.splitlines() if ln.strip()]
    for ln in lines:
        m = LINE.match(ln)
        if not m:
            continue
        qty, name, price = m.groups()
        if name.strip().upper().startswith("TOTAL"):
            total = float(price)
            continue
        items.append({"qty": int(qty or 1), "name": name.strip(),
                      "price": float(price)})
    if total is None:
        total = round(sum(it["qty"] * it["price"] for it in items), 2)
    return {"items": items, "total": total, "n_lines": len(lines)}


@app.function()
@modal.asgi_app(label="ocr")
def webapp():
    """Static frontend + job API in one ASGI app (reference splits these into
    a React build and an api mount; here the frontend is plain ES modules)."""
    from fastapi import FastAPI
    from fastapi.responses import FileResponse
    from fastapi.staticfiles import StaticFiles

    api = FastAPI()

    @api.post("/api/submit")
    def submit(payload: dict) -> dict:
        call = parse_receipt.spawn(payload.get("text", ""))
        return {"call_id": call.object_id}

    @api.get("/api/status")
    def status(call_id: str) -> dict:
        fc = modal.FunctionCall.from_id(call_id)
        try:
            return {"status": "done", "result": fc.get(timeout=0)}
        except TimeoutError:
            return {"status": "pending"}
        except Exception as e:  # job raised
            return {"status": "error", "detail": str(e)}

    @api.get("/")
    def index():
        return FileResponse(FRONTEND / "index.html")

    api.mount("/", StaticFiles(directory=FRONTEND), name="static")
    return api


RECEIPT = """CORNER COFFEE
2x latte $4.50
1 bagel $3.25
TOTAL $12.25
"""


@app.local_entrypoint()
def main():
    import asyncio
    import time

    import httpx

    asgi = webapp.raw()

    async def go():
        transport = httpx.ASGITransport(app=asgi)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://ocr") as c:
            page = await c.get("/")
            assert page.status_code == 200 and "Receipt parser" in page.text
            js = await c.get("/app.js")
            assert js.status_code == 200 and "api/submit" in js.text
            r = await c.post("/api/submit", json={"text": RECEIPT})
            call_id = r.json()["call_id"]
            deadline = time.time() + 30
            while True:
                s = (await c.get("/api/status",
                                 params={"call_id": call_id})).json()
                if s["status"] == "done":
                    return s["result"]
                assert s["status"] == "pending" and time.time() < deadline
                await asyncio.sleep(0.2)

    parsed = asyncio.run(go())
    print("parsed:", parsed)
    assert parsed["total"] == 12.25
    assert len(parsed["items"]) == 2
    print("frontend + job queue OK")
