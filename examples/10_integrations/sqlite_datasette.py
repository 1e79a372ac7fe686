# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/sqlite_datasette.py"]
# ---
# # Read-only SQL explorer over a Volume database (the datasette role)
#
# Datasette publishes a sqlite file as a queryable web API.  Same shape: a
# cron job refreshes a database on a Volume; a web endpoint executes
# READ-ONLY SQL against it (mutations rejected) and returns JSON rows.

import modal_examples_amd as modal

app = modal.App("example-sqlite-datasette")

data = modal.Volume.from_name("datasette-db", create_if_missing=True)


@app.function(schedule=modal.Period(hours=24))
def refresh_db():
    import sqlite3

    con = sqlite3.connect(data.path / "fleet.db")
    con.executescript("""
        DROP TABLE IF EXISTS gpu_runs;
        CREATE TABLE gpu_runs (day TEXT, model TEXT, images INTEGER);
        INSERT INTO gpu_runs VALUES
          ('2026-09-10','sdxl',1200),('2026-09-10','flux',300),
          ('2026-09-11','sdxl',1550),('2026-09-11','flux',410),
          ('2026-09-12','sdxl',1610),('2026-09-12','flux',502);
    """)
    con.commit()
    con.close()
    data.commit()


@app.function()
@modal.fastapi_endpoint(method="GET", label="sql")
def query(q: str = "SELECT model, SUM(images) AS total FROM gpu_runs GROUP BY model"):
    import sqlite3

    if not q.lstrip().lower().startswith("select"):
        return {"error": "read-only: SELECT queries only"}
    data.reload()
    con = sqlite3.connect(f"file:{data.path / 'fleet.db'}?mode=ro", uri=True)
    try:
        cur = con.execute(q)
        cols = [c[0] for c in cur.description]
        return {"columns": cols, "rows": [list(r) for r in cur.fetchall()]}
    except sqlite3.Error as e:
        return {"error": str(e)}
    finally:
        con.close()


@app.local_entrypoint()
def main():
    import asyncio

    import httpx

    from modal_examples_amd.web.ingress import build_ingress_app

    refresh_db.remote()

    async def go():
        root = build_ingress_app(app)
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            ok = (await c.get("/sql")).json()
            blocked = (await c.get("/sql", params={
                "q": "DROP TABLE gpu_runs"})).json()
            return ok, blocked

    ok, blocked = asyncio.run(go())
    assert ok["columns"] == ["model", "total"], ok
    assert dict(ok["rows"]) == {"flux": 1212, "sdxl": 4360}, ok
    assert "read-only" in blocked["error"]
    print("totals:", dict(ok["rows"]))
