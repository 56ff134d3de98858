"""The driver's bench.py contract: runs standalone, prints one JSON line
with the required keys (tiny CPU shape)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    env = dict(os.environ)
    env["NXDT_DISABLE_TUNABLEOP"] = "1"
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "1",
         "--warmup", "0", "--seq", "64", "--layers", "1", "--gbs", "2",
         "--mbs", "1"],
        capture_output=True, text=True, timeout=900, env=env, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["data"] == "synthetic"
    assert rec["config"]["model"] == "llama3-8B"


def test_bench_torchrun_world2_contract():
    """The driver's exact multi-rank invocation: torchrun x2 (gloo on CPU),
    TP2+SP+ZeRO-1, one JSON line from rank 0."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--seq", "64", "--layers", "2",
         "--gbs", "2", "--mbs", "1"],
        capture_output=True, text=True, timeout=900, cwd=repo,
    )
    assert r.returncode == 0, r.stderr[-3000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout[-2000:]
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "tp2_zero1_sp"
    assert rec["value"] > 0
