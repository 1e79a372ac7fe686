# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/long_training.py"]
# ---
# # Resumable long training
#
# The interruption-survival pattern: a short `timeout` kills the function
# mid-training, `Retries` restarts it, and each restart resumes from the
# checkpoint on the Volume — so the run converges across interruptions.

import time

import modal_examples_amd as modal

app = modal.App("example-long-training")

ckpts = modal.Volume.from_name("long-training-ckpts", create_if_missing=True)

TOTAL_STEPS = 6
STEPS_PER_LIFE = 2  # how far one container gets before the timeout kills it


@app.function(
    timeout=4,
    retries=modal.Retries(initial_delay=0.0, max_retries=10),
    single_use_containers=True,
)
def train() -> int:
    import json

    ck = ckpts.path / "state.json"
    step = json.loads(ck.read_text())["step"] if ck.exists() else 0
    print(f"resuming from step {step}")
    lives = 0
    while step < TOTAL_STEPS:
        time.sleep(0.3)  # one "training step"
        step += 1
        ck.write_text(json.dumps({"step": step}))
        ckpts.commit()
        lives += 1
        if lives >= STEPS_PER_LIFE and step < TOTAL_STEPS:
            print(f"simulating preemption at step {step}")
            time.sleep(60)  # blow past the timeout → killed → retried
    return step


@app.local_entrypoint()
def main():
    if (ckpts.path / "state.json").exists():
        (ckpts.path / "state.json").unlink()
    final = train.remote()
    print(f"training survived interruptions, finished at step {final}")
    assert final == TOTAL_STEPS
