# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/04_secrets/db_to_sheet.py"]
# ---
# # Scheduled ETL: database → report sheet
#
# The db-to-sheet pattern: a cron-scheduled function reads a database with
# credentials from a Secret and publishes an aggregate "sheet" (CSV on a
# Volume — the Google-Sheets stand-in in this network-free environment).

import sqlite3

import modal_examples_amd as modal

app = modal.App("example-db-to-sheet")

db_secret = modal.Secret.from_dict({"DB_PATH": "", "DB_TOKEN": "tok-local"})
reports = modal.Volume.from_name("etl-reports", create_if_missing=True)


def seed_database(path: str):
    conn = sqlite3.connect(path)
    conn.execute("CREATE TABLE IF NOT EXISTS orders (city TEXT, amount REAL)")
    conn.execute("DELETE FROM orders")
    conn.executemany("INSERT INTO orders VALUES (?,?)", [
        ("tokyo", 120.0), ("tokyo", 80.5), ("lima", 42.0), ("oslo", 99.9),
        ("lima", 13.1)])
    conn.commit()
    conn.close()


@app.function(secrets=[db_secret], schedule=modal.Cron("0 7 * * *"))
def etl() -> str:
    import os

    db_path = os.environ.get("DB_PATH") or str(reports.path / "source.db")
    assert os.environ["DB_TOKEN"], "credentials come from the Secret"
    conn = sqlite3.connect(db_path)
    rows = conn.execute(
        "SELECT city, SUM(amount), COUNT(*) FROM orders GROUP BY city"
        " ORDER BY 2 DESC").fetchall()
    conn.close()
    csv = "city,total,orders\n" + "\n".join(
        f"{c},{t:.2f},{n}" for c, t, n in rows)
    (reports.path / "daily_report.csv").write_text(csv)
    reports.commit()
    return csv


@app.local_entrypoint()
def main():
    seed_database(str(reports.path / "source.db"))
    print(etl.remote())
    assert "tokyo,200.50,2" in (reports.path / "daily_report.csv").read_text()
