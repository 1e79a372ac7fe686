# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/08_advanced/generators_async.py"]
# ---
# # Async generators (08_advanced/generators_async.py role): stream results
# # from a remote generator with `async for`.

import modal_examples_amd as modal

app = modal.App("example-generators-async")


@app.function()
def countdown(n: int):
    import time

    for i in range(n, 0, -1):
        time.sleep(0.05)
        yield i


@app.local_entrypoint()
def main():
    import asyncio

    async def go():
        seen = []
        async for item in countdown.remote_gen.aio(5):
            seen.append(item)
        return seen

    seen = asyncio.run(go())
    assert seen == [5, 4, 3, 2, 1]
    print("streamed:", seen)
