# ---
# deploy: true
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/say_hello_cron.py"]
# ---
# # A deployed cron greeter (misc/say_hello_cron.py role).

import time

import modal_examples_amd as modal

app = modal.App("example-say-hello-cron")

greetings = modal.Dict.from_name("cron-greetings", create_if_missing=True)


@app.function(schedule=modal.Period(seconds=1))
def say_hello():
    n = (greetings.get("count") or 0) + 1
    greetings.put("count", n)
    print(f"hello #{n} at {time.strftime('%H:%M:%S')}")


@app.local_entrypoint()
def main():
    greetings.put("count", 0)
    with modal.enable_output():
        app.deploy()
        deadline = time.time() + 15
        while time.time() < deadline and (greetings.get("count") or 0) < 2:
            time.sleep(0.5)
    fired = greetings.get("count") or 0
    assert fired >= 2, f"cron fired {fired} times"
    print(f"cron fired {fired} times in the window")
