"""e2e A/B harness: run bench.py under different env knob combos, report
median-ish (bench itself amortizes over steps)."""
import json, os, subprocess, sys
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CONFIGS = [
    ("baseline", {}),
    ("midt7", {"PIPEEDGE_GEMM_MIDT": "7"}),
    ("midt10", {"PIPEEDGE_GEMM_MIDT": "10"}),
    ("midt7+kvt64", {"PIPEEDGE_GEMM_MIDT": "7", "PIPEEDGE_ATTN_V2": "3"}),
    ("streams3", {"PIPEEDGE_UBATCH_STREAMS": "3"}),
    ("midt7+streams3", {"PIPEEDGE_GEMM_MIDT": "7",
                        "PIPEEDGE_UBATCH_STREAMS": "3"}),
]

def run(env_extra, steps=60):
    env = dict(os.environ, **env_extra)
    r = subprocess.run([sys.executable, os.path.join(REPO, "bench.py"),
                        "--steps", str(steps), "--warmup", "8"],
                       capture_output=True, text=True, env=env, cwd=REPO)
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)["value"]

for name, env in CONFIGS:
    vals = [run(env) for _ in range(2)]
    print(f"{name:18s} {vals[0]:9.1f} {vals[1]:9.1f} img/s")
