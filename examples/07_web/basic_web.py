# ---
# deploy: true
# cmd: ["python", "-m", "modal_examples_amd", "serve", "examples/07_web/basic_web.py", "--timeout", "3"]
# ---
# # Web endpoints
#
# `@modal.fastapi_endpoint` turns functions into HTTP routes on the local
# ingress: GET with query params, POST with a JSON body, and a class-based
# endpoint whose model loads once in `@modal.enter`.

import modal_examples_amd as modal

app = modal.App("example-basic-web")


@app.function()
@modal.fastapi_endpoint(method="GET", label="square", docs=True)
def square(x: int = 2) -> dict:
    return {"x": x, "square": x * x}


@app.function()
@modal.fastapi_endpoint(method="POST", label="goodbye")
def goodbye(name: str = "world") -> dict:
    return {"farewell": f"goodbye, {name}"}


@app.function()
@modal.fastapi_endpoint(method="GET", label="locked", requires_proxy_auth=True)
def locked() -> dict:
    return {"secret": "only with proxy auth"}


@app.cls()
class Greeter:
    @modal.enter()
    def setup(self):
        self.greeting = "hello from a warm container"

    @modal.fastapi_endpoint(method="GET", label="greet")
    def greet(self, name: str = "you") -> dict:
        return {"message": f"{self.greeting}, {name}"}


@app.local_entrypoint()
def main():
    import httpx

    from modal_examples_amd.web.ingress import serve, stop_serving

    base = serve(app, port=8794)
    try:
        r = httpx.get(f"{base}/square", params={"x": 7}, timeout=30)
        print(r.json())
        assert r.json()["square"] == 49
        r2 = httpx.post(f"{base}/goodbye", json={"name": "moon"}, timeout=30)
        print(r2.json())
        r3 = httpx.get(f"{base}/greet", params={"name": "tester"}, timeout=30)
        print(r3.json())
    finally:
        stop_serving()
