# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/08_advanced/poll_delayed_result.py"]
# ---
# Fire-and-forget: spawn a call, keep only its id, poll for the result later
# (from any process — results are durable in the local store).

import time

import modal_examples_amd as modal

app = modal.App("example-poll")


@app.function()
def slow_job(x: int) -> int:
    time.sleep(1.0)
    return x * 10


@app.local_entrypoint()
def main():
    call_id = slow_job.spawn(7).object_id
    print("spawned", call_id)

    fc = modal.FunctionCall.from_id(call_id)
    while True:
        try:
            result = fc.get(timeout=0)
            break
        except TimeoutError:
            print("still running...")
            time.sleep(0.3)
    print("result:", result)
    assert result == 70
