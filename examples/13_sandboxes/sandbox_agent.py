# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/13_sandboxes/sandbox_agent.py"]
# ---
# # An agent loop over a Sandbox
#
# The sandbox-agent shape (reference: 13_sandboxes/sandbox_agent.py — an LLM
# choosing shell commands, executing them in a Sandbox, reading the output):
# here the policy is a random-init GPT scoring the candidate actions (so the
# loop is hermetic — no API keys), with a repeat-memory so exploration
# terminates.  The mechanics are the real thing: propose → exec in sandbox →
# observe → update context → repeat until the goal check passes.

import modal_examples_amd as modal

app = modal.App("example-sandbox-agent")

N_FILES = 6
GOAL_MARKER = "FLAG{mi355x}"


@app.function(gpu="mi355x", timeout=600)
def run_agent(max_steps: int = 12) -> dict:
    import torch

    from modal_examples_amd.models.gpt.model import GPT, GPTConfig

    # --- the environment: a sandbox with files, one contains the marker
    sb = modal.Sandbox.create("sleep", "600")
    import pathlib

    root = pathlib.Path(sb.workdir)
    target = 4  # deterministic for the self-test
    for i in range(N_FILES):
        body = GOAL_MARKER if i == target else f"nothing here ({i})"
        (root / f"note_{i}.txt").write_text(body + "\n")

    # --- the policy: GPT logits over the action space
    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=256, block_size=128, n_layer=2, n_head=2,
                    n_embd=64)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    policy = GPT(cfg).to(device)
    actions = [f"cat note_{i}.txt" for i in range(N_FILES)]

    def choose(context: str, tried: set) -> int:
        ids = torch.tensor([[b % 256 for b in context.encode()[-96:]]],
                           device=device)
        with torch.no_grad():
            logits, _ = policy(ids)
        scores = logits[0, -1, : len(actions)].clone()
        for t in tried:  # repeat-memory: never re-run an action
            scores[t] = float("-inf")
        return int(scores.argmax())

    # --- the loop
    context = "goal: find the file containing the flag\n"
    tried: set = set()
    trace = []
    found = None
    for step in range(max_steps):
        a = choose(context, tried)
        tried.add(a)
        p = sb.exec("sh", "-c", actions[a])
        p.wait()
        obs = p.stdout.read().strip()
        trace.append({"step": step, "action": actions[a], "obs": obs[:40]})
        context += f"$ {actions[a]}\n{obs}\n"
        if GOAL_MARKER in obs:
            found = actions[a]
            break
    sb.terminate()
    return {"found_with": found, "steps": len(trace), "trace": trace}


@app.local_entrypoint()
def main():
    out = run_agent.remote()
    for t in out["trace"]:
        print(f"  step {t['step']}: {t['action']:18s} → {t['obs']}")
    assert out["found_with"] is not None, "agent never found the flag"
    assert out["steps"] <= N_FILES  # repeat-memory bounds the search
    print(f"agent found the flag via {out['found_with']!r} "
          f"in {out['steps']} steps")
