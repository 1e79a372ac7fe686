# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/03_scaling_out/dynamic_batching.py"]
# ---
# `@modal.batched` collects individual calls into one list-shaped call —
# on plain functions and on class methods (the GPU-side batcher that feeds
# batched kernel launches).

import modal_examples_amd as modal

app = modal.App("example-dynamic-batching")


@app.function()
@modal.batched(max_batch_size=4, wait_ms=500)
def batch_square(xs: list[int]) -> list[int]:
    print(f"batched call of size {len(xs)}")
    return [x * x for x in xs]


@app.cls()
class Doubler:
    @modal.enter()
    def setup(self):
        self.factor = 2

    @modal.batched(max_batch_size=4, wait_ms=500)
    def mul(self, xs: list[int]) -> list[int]:
        return [x * self.factor for x in xs]


@app.local_entrypoint()
def main():
    squares = list(batch_square.map(range(10)))
    assert squares == [x * x for x in range(10)]
    print("squares ok")
    doubles = list(Doubler().mul.map(range(10)))
    assert doubles == [x * 2 for x in range(10)]
    print("doubles ok")
