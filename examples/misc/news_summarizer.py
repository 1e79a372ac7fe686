# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/news_summarizer.py"]
# ---
# # Scheduled news digest (misc/news_summarizer.py role)
#
# The daily-digest shape: a Period schedule scans an inbox Volume for new
# articles, summarizes each (extractive: centrality-ranked sentences), and
# publishes the digest to a Dict the web/notification side reads.  The
# entrypoint seeds articles, deploys the schedule, and verifies the digest
# appears within the window.

import modal_examples_amd as modal

app = modal.App("example-news-summarizer")

inbox = modal.Volume.from_name("news-inbox", create_if_missing=True)
digests = modal.Dict.from_name("news-digests", create_if_missing=True)


def summarize(text: str, k: int = 2) -> str:
    """Extractive: pick the k most central sentences (bag-of-words cosine)."""
    import math
    import re
    from collections import Counter

    sents = [s.strip() for s in re.split(r"(?<=[.!?])\s+", text) if s.strip()]
    if len(sents) <= k:
        return " ".join(sents)
    bags = [Counter(re.findall(r"[a-z']+", s.lower())) for s in sents]

    def cos(a, b):
        num = sum(a[w] * b.get(w, 0) for w in a)
        den = math.sqrt(sum(v * v for v in a.values())) * \
            math.sqrt(sum(v * v for v in b.values())) or 1.0
        return num / den

    central = [(sum(cos(b, o) for o in bags) , i) for i, b in enumerate(bags)]
    keep = sorted(i for _, i in sorted(central, reverse=True)[:k])
    return " ".join(sents[i] for i in keep)


@app.function(schedule=modal.Period(seconds=1))
def build_digest():
    entries = {}
    for f in sorted(inbox.path.glob("*.txt")):
        entries[f.stem] = summarize(f.read_text())
    if entries:
        digests.put("latest", {"count": len(entries), "items": entries})


@app.local_entrypoint()
def main():
    import time

    digests.delete("latest")
    (inbox.path / "gpus.txt").write_text(
        "Accelerators keep getting faster. The MI355X offers 288 GB of HBM3E "
        "per device. Cooling remains a challenge. Vendors promise more. "
        "The MI355X pairs that memory with eight XCD chiplets.")
    (inbox.path / "markets.txt").write_text(
        "Markets rose today. Chip stocks led the gains. Analysts cite "
        "demand for inference capacity. Weather was sunny.")
    inbox.commit()
    with modal.enable_output():
        app.deploy()
        deadline = time.time() + 15
        latest = None
        while time.time() < deadline:
            latest = digests.get("latest")
            if latest and latest["count"] == 2:
                break
            time.sleep(0.5)
    assert latest and latest["count"] == 2, latest
    for name, summ in latest["items"].items():
        assert 0 < len(summ) < 200
        print(f"[{name}] {summ}")
