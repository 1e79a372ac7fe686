# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/09_job_queues/web_job_queue_wrapper.py"]
# ---
# A REST wrapper around spawn/status/result: POST a job, poll its id.
# (The OCR-job-queue pattern; here the "job" is a checksum.)

import hashlib
import time

import modal_examples_amd as modal

app = modal.App("example-web-job-queue")


@app.function()
def process_document(text: str) -> str:
    time.sleep(0.5)  # pretend to OCR
    return hashlib.sha256(text.encode()).hexdigest()[:16]


@app.function()
@modal.fastapi_endpoint(method="POST", label="submit")
def submit(text: str) -> dict:
    call = process_document.spawn(text)
    return {"call_id": call.object_id}


@app.function()
@modal.fastapi_endpoint(method="GET", label="result")
def result(call_id: str) -> dict:
    fc = modal.FunctionCall.from_id(call_id)
    try:
        return {"status": "done", "result": fc.get(timeout=0)}
    except TimeoutError:
        return {"status": "pending"}


@app.local_entrypoint()
def main():
    # exercise the same flow the endpoints wrap
    call = process_document.spawn("hello job queue")
    fc = modal.FunctionCall.from_id(call.object_id)
    while True:
        try:
            out = fc.get(timeout=0)
            break
        except TimeoutError:
            time.sleep(0.2)
    print("job result:", out)
