# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/dbt_pipeline.py"]
# ---
# # Staged SQL transformations with tests (the dbt role)
#
# The dbt shape: declarative MODELS (named SQL selects) build staging and
# mart tables in dependency order, then data TESTS assert invariants —
# executed by a Modal function against a warehouse database on a Volume.

import modal_examples_amd as modal

app = modal.App("example-dbt-pipeline")

warehouse = modal.Volume.from_name("dbt-warehouse", create_if_missing=True)

MODELS = [  # (name, SQL) in dependency order — the dbt DAG
    ("stg_orders",
     "SELECT id, customer, CAST(amount AS REAL) AS amount, status "
     "FROM raw_orders WHERE status != 'cancelled'"),
    ("stg_customers",
     "SELECT id, UPPER(region) AS region FROM raw_customers"),
    ("mart_revenue_by_region",
     "SELECT c.region, ROUND(SUM(o.amount), 2) AS revenue, COUNT(*) AS orders "
     "FROM stg_orders o JOIN stg_customers c ON o.customer = c.id "
     "GROUP BY c.region"),
]

TESTS = [  # (name, SQL returning VIOLATION rows) — dbt test semantics
    ("no_negative_revenue",
     "SELECT * FROM mart_revenue_by_region WHERE revenue < 0"),
    ("stg_orders_not_cancelled",
     "SELECT * FROM stg_orders WHERE status = 'cancelled'"),
    ("regions_unique",
     "SELECT region, COUNT(*) c FROM mart_revenue_by_region "
     "GROUP BY region HAVING c > 1"),
]


@app.function()
def seed() -> int:
    import sqlite3

    con = sqlite3.connect(warehouse.path / "wh.db")
    con.executescript("""
        DROP TABLE IF EXISTS raw_orders; DROP TABLE IF EXISTS raw_customers;
        CREATE TABLE raw_orders (id, customer, amount, status);
        CREATE TABLE raw_customers (id, region);
        INSERT INTO raw_orders VALUES
          (1, 10, '120.50', 'paid'), (2, 11, '80.00', 'paid'),
          (3, 10, '15.25', 'cancelled'), (4, 12, '990.10', 'paid'),
          (5, 11, '45.00', 'refunded');
        INSERT INTO raw_customers VALUES
          (10, 'emea'), (11, 'amer'), (12, 'apac');
    """)
    con.commit()
    con.close()
    warehouse.commit()
    return 5


@app.function()
def run_models() -> list:
    import sqlite3

    warehouse.reload()
    con = sqlite3.connect(warehouse.path / "wh.db")
    built = []
    for name, sql in MODELS:
        con.executescript(
            f"DROP TABLE IF EXISTS {name}; CREATE TABLE {name} AS {sql};")
        n = con.execute(f"SELECT COUNT(*) FROM {name}").fetchone()[0]
        built.append({"model": name, "rows": n})
    con.commit()
    con.close()
    warehouse.commit()
    return built


@app.function()
def run_tests() -> list:
    import sqlite3

    warehouse.reload()
    con = sqlite3.connect(warehouse.path / "wh.db")
    out = []
    for name, sql in TESTS:
        violations = con.execute(sql).fetchall()
        out.append({"test": name, "passed": not violations,
                    "violations": len(violations)})
    con.close()
    return out


@app.local_entrypoint()
def main():
    print("seeded", seed.remote(), "raw orders")
    for m in run_models.remote():
        print(f"built {m['model']}: {m['rows']} rows")
    results = run_tests.remote()
    for t in results:
        print(f"test {t['test']}: {'PASS' if t['passed'] else 'FAIL'}")
    assert all(t["passed"] for t in results), results
