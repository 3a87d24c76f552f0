import json, os, subprocess, sys
repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
for extra in ([], ["--episode-length", "200"], ["--policy", "mlp64"], ["--policy", "mlp64", "--episode-length", "200"]):
    out = subprocess.run([sys.executable, os.path.join(repo, "bench.py"), "--steps", "15", "--warmup", "3", *extra],
                         capture_output=True, text=True, timeout=300)
    d = json.loads(out.stdout.strip().splitlines()[-1])
    print(f"{d['metric'][:44]:44s} T={d['config']['seq_len']:5d}  {d['value']:>12,.0f} sol/s  {d['ms_per_step']:7.2f} ms")
