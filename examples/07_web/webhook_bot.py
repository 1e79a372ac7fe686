# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/webhook_bot.py"]
# ---
# # A chat-ops webhook bot (the 07_web/discord_bot role)
#
# The Discord-bot shape without the Discord SDK: the platform POSTs events
# to our webhook; slow work is ACKed immediately and `spawn`ed to run in the
# background; the bot then posts its answer back to the platform's callback
# URL (here: a second endpoint standing in for the chat service).

import modal_examples_amd as modal

app = modal.App("example-webhook-bot")

outbox = modal.Dict.from_name("bot-outbox", create_if_missing=True)


@app.function()
def handle_command(channel: str, command: str, arg: str) -> str:
    """The slow bot work (runs detached from the webhook request)."""
    import time

    if command == "/roll":
        import random

        n = int(arg or 20)
        reply = f"🎲 you rolled {random.Random(channel).randint(1, n)} / {n}"
    elif command == "/status":
        time.sleep(0.2)  # pretend to check the fleet
        reply = "all 8 GPUs healthy"
    else:
        reply = f"unknown command {command}"
    # post back to the chat service's callback (our stand-in endpoint)
    outbox.put(channel, reply)
    return reply


@app.function()
@modal.fastapi_endpoint(method="POST", label="webhook")
def webhook(channel: str = "", command: str = "", arg: str = ""):
    """Platforms demand an ACK within seconds: spawn and return.  The
    endpoint runs inside a worker container, so it hands off BY NAME (the
    store-backed dispatcher pattern, 09_job_queues/pipeline_orchestration)."""
    bot = modal.Function.from_name("example-webhook-bot", "handle_command")
    call = bot.spawn(channel, command, arg)
    return {"ack": True, "job": call.object_id}


@app.function()
@modal.fastapi_endpoint(method="GET", label="channel")
def read_channel(channel: str = ""):
    return {"channel": channel, "last_message": outbox.get(channel)}


@app.local_entrypoint()
def main():
    import asyncio
    import time

    import httpx

    from modal_examples_amd.web.ingress import build_ingress_app

    async def go():
        root = build_ingress_app(app)
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            t0 = time.monotonic()
            ack = (await c.post("/webhook", json={
                "channel": "ops", "command": "/status"})).json()
            ack_ms = (time.monotonic() - t0) * 1000
            assert ack["ack"] and ack["job"].startswith("fc-")
            deadline = time.monotonic() + 20
            msg = None
            while time.monotonic() < deadline:
                msg = (await c.get("/channel", params={"channel": "ops"})
                       ).json()["last_message"]
                if msg:
                    break
                await asyncio.sleep(0.2)
            return ack_ms, msg

    ack_ms, msg = asyncio.run(go())
    assert msg == "all 8 GPUs healthy", msg
    print(f"webhook ACKed in {ack_ms:.0f} ms; bot replied: {msg!r}")
