# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/08_advanced/hello_world_async.py"]
# ---
# # Async verbs (08_advanced/hello_world_async.py role): `.remote.aio` and
# # `.map.aio` from an async entrypoint, with concurrent awaits.

import modal_examples_amd as modal

app = modal.App("example-hello-async")


@app.function()
def slow_double(x: int) -> int:
    import time

    time.sleep(0.2)
    return 2 * x


@app.local_entrypoint()
def main():
    import asyncio
    import time

    async def go():
        t0 = time.monotonic()
        a, b, c = await asyncio.gather(
            slow_double.remote.aio(1),
            slow_double.remote.aio(2),
            slow_double.remote.aio(3),
        )
        concurrent_s = time.monotonic() - t0
        out = []
        async for r in slow_double.map.aio(range(6)):
            out.append(r)
        return (a, b, c), concurrent_s, out

    firsts, dt, mapped = asyncio.run(go())
    assert firsts == (2, 4, 6) and mapped == [0, 2, 4, 6, 8, 10]
    print(f"3 concurrent awaits in {dt:.2f}s (serial would be >=0.6s); "
          f"map.aio -> {mapped}")
