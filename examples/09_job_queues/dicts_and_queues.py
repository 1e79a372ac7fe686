# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/09_job_queues/dicts_and_queues.py"]
# ---
# Distributed state: a breadth-first "crawler" over a synthetic link graph.
# A shared `Queue` holds the frontier, a shared `Dict` deduplicates —
# multiple workers drain the queue concurrently.

import modal_examples_amd as modal

app = modal.App("example-dicts-queues")

frontier = modal.Queue.from_name("crawl-frontier", create_if_missing=True)
seen = modal.Dict.from_name("crawl-seen", create_if_missing=True)


def links_of(page: int) -> list[int]:
    # synthetic deterministic link graph over 50 pages
    return [(page * 7 + k) % 50 for k in (1, 2, 3)]


@app.function()
def crawl_worker(worker_id: int) -> int:
    q = modal.Queue.from_name("crawl-frontier")
    d = modal.Dict.from_name("crawl-seen")
    from queue import Empty

    crawled = 0
    while True:
        try:
            page = q.get(block=True, timeout=1.0)
        except Empty:
            break
        if page is None:
            break
        if not d.put_if_absent(page, worker_id):
            continue  # another worker claimed this page (atomic dedup)
        crawled += 1
        q.put_many([p for p in links_of(page) if not d.contains(p)])
    return crawled


@app.local_entrypoint()
def main():
    frontier.clear(all=True)
    seen.clear()
    frontier.put(0)
    counts = list(crawl_worker.map(range(4)))
    print(f"workers crawled {counts} (total {sum(counts)} pages, "
          f"{seen.len()} unique)")
    assert sum(counts) == seen.len()
    seen.clear()
