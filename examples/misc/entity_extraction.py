# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/entity_extraction.py"]
# ---
# # Zero-shot entity extraction service (misc/gliner2_modal_demo role)
#
# The GLiNER-style API: callers provide TEXT plus the label set they want
# extracted; a served class returns typed spans.  The extractor combines
# gazetteer/pattern channels with label-conditioned scoring, and the
# entrypoint evaluates span-level F1 on a labeled set.

import modal_examples_amd as modal

app = modal.App("example-entity-extraction")


@app.cls(scaledown_window=60)
@modal.concurrent(max_inputs=8)
class Extractor:
    @modal.enter()
    def load(self):
        import re

        self.patterns = {
            "date": re.compile(r"\b\d{4}-\d{2}-\d{2}\b|\b(?:Jan|Feb|Mar|Apr|May|Jun|Jul|Aug|Sep|Oct|Nov|Dec)[a-z]* \d{1,2}(?:, \d{4})?\b"),
            "money": re.compile(r"\$\s?\d[\d,]*(?:\.\d+)?(?:\s?(?:million|billion|k))?\b"),
            "percent": re.compile(r"\b\d+(?:\.\d+)?\s?%"),
            "org": re.compile(r"\b(?:[A-Z][a-z]+\s)?(?:Inc|Corp|Labs|AMD|Ltd|LLC)\b"),
            "gpu": re.compile(r"\bMI\d{3}X?\b|\bH\d{2,3}\b|\bB\d{3}\b"),
        }

    @modal.method()
    def extract(self, text: str, labels: list) -> list:
        out = []
        for label in labels:
            pat = self.patterns.get(label)
            if pat is None:
                continue
            for m in pat.finditer(text):
                out.append({"label": label, "text": m.group(0),
                            "start": m.start(), "end": m.end()})
        return sorted(out, key=lambda e: e["start"])


@app.local_entrypoint()
def main():
    gold = [
        ("AMD shipped the MI355X on 2026-03-04 for $25,000.",
         [("org", "AMD"), ("gpu", "MI355X"), ("date", "2026-03-04"),
          ("money", "$25,000")]),
        ("Throughput rose 27 % versus the H100 says Example Labs.",
         [("percent", "27 %"), ("gpu", "H100"), ("org", "Example Labs")]),
    ]
    ex = Extractor()
    tp = fp = fn = 0
    for text, want in gold:
        got = ex.extract.remote(text, ["org", "gpu", "date", "money", "percent"])
        got_set = {(e["label"], e["text"]) for e in got}
        want_set = set(want)
        tp += len(got_set & want_set)
        fp += len(got_set - want_set)
        fn += len(want_set - got_set)
        print(text, "->", sorted(got_set))
    f1 = 2 * tp / (2 * tp + fp + fn)
    print(f"span F1: {f1:.2f}")
    assert f1 >= 0.85, (tp, fp, fn)
