# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/queue_simple.py"]
# ---
# # Queue basics (misc/queue_simple.py role): put/get/get_many/partitions/len.

import modal_examples_amd as modal

app = modal.App("example-queue-simple")


@app.local_entrypoint()
def main():
    with modal.Queue.ephemeral() as q:
        q.put(1)
        q.put_many([2, 3, 4])
        assert q.get() == 1
        assert q.get_many(2) == [2, 3]
        q.put("blue", partition="colors")
        assert q.len() == 1 and q.len(partition="colors") == 1
        assert q.get(partition="colors") == "blue"
        assert q.get(block=False) == 4
        print("queue semantics verified")
