# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/05_scheduling/schedule_simple.py"]
# ---
# Scheduled functions: `modal.Period` intervals and `modal.Cron` expressions
# fire after `deploy`.  Running this entrypoint demonstrates one manual fire.

import time

import modal_examples_amd as modal

app = modal.App("example-schedules")


@app.function(schedule=modal.Period(seconds=5))
def heartbeat():
    print("heartbeat at", time.strftime("%T"))
    d = modal.Dict.from_name("heartbeat-log", create_if_missing=True)
    d[time.time()] = "beat"


@app.function(schedule=modal.Cron("0 9 * * 1-5"))
def weekday_report():
    print("good morning — weekday report")


@app.local_entrypoint()
def main():
    # schedules run under `deploy`; here we call once to show they're plain
    # functions too
    heartbeat.remote()
    print("heartbeat fired once; `python -m modal_examples_amd deploy` "
          "keeps firing it every 5s")
    modal.Dict.from_name("heartbeat-log").clear()
