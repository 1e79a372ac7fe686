# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/09_job_queues/doc_ocr_jobs.py"]
# ---
# # Document OCR job queue — backend (09_job_queues/doc_ocr_jobs.py role)
#
# The spawn-based job backend the webapp fronts: submit parse jobs with
# `.spawn` (durable FunctionCall ids), retries=3 on the worker function,
# poll by id from anywhere.  The OCR stand-in extracts digit-strings from
# synthetic "documents".

import modal_examples_amd as modal

app = modal.App("example-doc-ocr-jobs")


@app.function(retries=3, timeout=120)
def parse_receipt(doc: str) -> dict:
    """The OCR role: pull amounts out of the scanned text."""
    import re

    amounts = [float(m) for m in re.findall(r"\d+\.\d{2}", doc)]
    return {"lines": len(doc.splitlines()), "amounts": amounts,
            "total": round(sum(amounts), 2)}


@app.local_entrypoint()
def main():
    docs = [
        "COFFEE 3.50\nBAGEL 2.25\nTOTAL 5.75",
        "TAXI 23.40\nTIP 4.00",
        "BOOK 15.99",
    ]
    # submit: durable job ids a separate process could poll
    calls = [parse_receipt.spawn(d) for d in docs]
    ids = [c.object_id for c in calls]
    print("submitted jobs:", ids)
    # poll by id (the webapp's status endpoint does exactly this)
    import modal_examples_amd as modal_

    results = [modal_.FunctionCall.from_id(i).get(timeout=60) for i in ids]
    totals = [r["total"] for r in results]
    assert totals == [11.5, 27.4, 15.99], totals
    print("totals:", totals)
