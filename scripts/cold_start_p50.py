import json
import subprocess
import sys

samples = []
for i in range(5):
    r = subprocess.run([sys.executable, "bench.py", "--steps", "1",
                        "--warmup", "1", "--batch", "4"],
                       capture_output=True, text=True, timeout=240)
    for line in r.stdout.splitlines():
        if line.startswith("{"):
            d = json.loads(line)
            samples.append(d["cold_start_s"])
            print(f"sample {i}: cold_start={d['cold_start_s']}s "
                  f"value={d['value']}")
samples.sort()
print(json.dumps({"p50_cold_start_s": samples[len(samples) // 2],
                  "samples": samples}))
