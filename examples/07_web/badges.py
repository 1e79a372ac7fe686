# ---
# deploy: true
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/badges.py"]
# ---
# # Dynamic SVG badge endpoint (07_web/badges.py role)
#
# A deployable endpoint rendering shields.io-style SVG badges with live data
# (here: the app's own call counter from a shared Dict).

import modal_examples_amd as modal

app = modal.App("example-badges")

stats = modal.Dict.from_name("badge-stats", create_if_missing=True)


def _badge_svg(label: str, value: str, color: str = "#4c1") -> str:
    lw, vw = 6 * len(label) + 12, 6 * len(value) + 12
    return f"""<svg xmlns="http://www.w3.org/2000/svg" width="{lw+vw}" height="20">
<rect width="{lw}" height="20" fill="#555"/>
<rect x="{lw}" width="{vw}" height="20" fill="{color}"/>
<g fill="#fff" font-family="Verdana" font-size="11">
<text x="{lw/2}" y="14" text-anchor="middle">{label}</text>
<text x="{lw+vw/2}" y="14" text-anchor="middle">{value}</text>
</g></svg>"""


@app.function()
@modal.fastapi_endpoint(method="GET", label="badge")
def badge(label: str = "runs", color: str = "#4c1"):
    from fastapi.responses import Response

    n = (stats.get("hits") or 0) + 1
    stats.put("hits", n)
    return Response(content=_badge_svg(label, str(n), color),
                    media_type="image/svg+xml")


@app.local_entrypoint()
def main():
    import asyncio

    import httpx

    from modal_examples_amd.web.ingress import build_ingress_app

    stats.put("hits", 0)

    async def go():
        root = build_ingress_app(app)
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            r1 = await c.get("/badge", params={"label": "ci"})
            r2 = await c.get("/badge", params={"label": "ci"})
            assert r1.headers["content-type"].startswith("image/svg")
            assert ">1<" in r1.text and ">2<" in r2.text
            return r2

    asyncio.run(go())
    print("badge endpoint serves live SVG counters")
