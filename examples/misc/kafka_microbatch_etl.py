# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/kafka_microbatch_etl.py"]
# ---
# # Micro-batch streaming ETL (misc/kafka_microbatch_etl.py role)
#
# The Kafka-consumer shape without Kafka: producers append events to a
# partitioned `modal.Queue` topic; a consumer function drains micro-batches
# with `get_many`, aggregates per window, and commits results to a Dict —
# offsets semantics via the queue's FIFO claims (each event delivered once).

import modal_examples_amd as modal

app = modal.App("example-kafka-etl")

topic = modal.Queue.from_name("etl-topic", create_if_missing=True)
aggregates = modal.Dict.from_name("etl-aggregates", create_if_missing=True)


@app.function()
def produce(partition: str, n: int) -> int:
    import random

    random.seed(hash(partition) % 1000)
    for i in range(n):
        topic.put({"user": partition, "amount": random.randint(1, 9)},
                  partition=partition)
    return n


@app.function()
def consume(partition: str, batch: int = 16) -> dict:
    total = count = batches = 0
    while True:
        events = topic.get_many(batch, partition=partition, block=False)
        if not events:
            break
        batches += 1
        for e in events:
            total += e["amount"]
            count += 1
    out = {"partition": partition, "events": count, "sum": total,
           "micro_batches": batches}
    aggregates.put(partition, out)
    return out


@app.local_entrypoint()
def main():
    parts = ["alpha", "beta", "gamma"]
    produced = list(produce.starmap([(p, 40) for p in parts]))
    results = list(consume.map(parts))
    for r in results:
        assert r["events"] == 40, r  # exactly-once drain per partition
        assert r["micro_batches"] >= 3
    assert sum(produced) == sum(r["events"] for r in results)
    print("aggregates:", {r["partition"]: r["sum"] for r in results})
