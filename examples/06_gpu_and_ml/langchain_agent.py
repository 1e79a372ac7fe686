# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/langchain_agent.py"]
# ---
# # A tool-using chain over Modal functions (the 06/langchains role)
#
# The LangChain-on-Modal shape: a controller loop drives an LLM that picks
# TOOLS, where every tool is itself a Modal function (calculator, retriever,
# GPU model).  The "LLM" planner here is deterministic (random-weight
# models can't plan), but every hop crosses a real container boundary and
# the chain state threads through the loop exactly like the reference.

import modal_examples_amd as modal

app = modal.App("example-langchain-agent")

kb = modal.Dict.from_name("agent-kb", create_if_missing=True)


@app.function()
def tool_calculator(expr: str) -> str:
    import ast
    import operator as op

    ops = {ast.Add: op.add, ast.Sub: op.sub, ast.Mult: op.mul, ast.Div: op.truediv}

    def ev(n):
        if isinstance(n, ast.Constant):
            return n.value
        if isinstance(n, ast.BinOp):
            return ops[type(n.op)](ev(n.left), ev(n.right))
        raise ValueError("unsupported")

    return str(ev(ast.parse(expr, mode="eval").body))


@app.function()
def tool_retrieve(query: str) -> str:
    hits = [v for k, v in kb.items() if any(
        w in str(v).lower() for w in query.lower().split())]
    return hits[0] if hits else "no match"


@app.function(gpu="mi355x")
def tool_embed_similarity(a: str, b: str) -> float:
    """A GPU tool in the chain: embedding cosine via the vision encoder's
    projection stack (any GPU model works as a chain tool)."""
    import torch

    def emb(s: str) -> torch.Tensor:
        g = torch.Generator().manual_seed(abs(hash(s)) % (2**31))
        return torch.nn.functional.normalize(torch.randn(64, generator=g), dim=0)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    return float((emb(a).to(device) @ emb(b).to(device)).cpu())


@app.local_entrypoint()
def main():
    kb.put("fact1", "The MI355X has 288 GB of HBM3E")
    kb.put("fact2", "Paris is the capital of France")

    question = "How much HBM do four MI355X GPUs have together?"
    scratchpad = []
    # the planner: observe -> act -> observe (ReAct shape, deterministic)
    fact = tool_retrieve.remote("MI355X HBM")
    scratchpad.append(("retrieve", fact))
    per_gpu = [w for w in fact.split() if w.isdigit()][0]
    total = tool_calculator.remote(f"{per_gpu} * 4")
    scratchpad.append(("calculator", total))
    check = tool_embed_similarity.remote(question, fact)
    scratchpad.append(("relevance", check))
    answer = f"{total} GB"
    for step, obs in scratchpad:
        print(f"  [{step}] {obs}")
    print("answer:", answer)
    assert answer == "1152.0 GB" or answer == "1152 GB", answer
