"""Driver-contract tests for bench.py: the exact torchrun launch the driver
uses must work (VERDICT r1 #1 — the multi-GPU path must be correct by
construction; on CPU it rides gloo with the same code path)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _parse_json_line(out: str) -> dict:
    for line in out.splitlines():
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out}")


def test_bench_single_process_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny_proxy", "--steps", "2",
         "--warmup", "1", "--mbs", "2", "--seq-len", "128"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    rec = _parse_json_line(r.stdout)
    assert rec["n_gpus"] == 1 and rec["steps"] == 2
    assert rec["config"]["loss_step0"] is not None


def test_bench_torchrun_world2_cpu():
    """The driver's N>1 launch shape: torch.distributed.run --nproc-per-node 2
    with RANK/WORLD_SIZE from env; FSDP2 over gloo; rank 0 prints ONE JSON."""
    env = dict(os.environ)
    env.pop("RANK", None); env.pop("WORLD_SIZE", None); env.pop("LOCAL_RANK", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29617", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--model", "tiny_proxy", "--mbs", "2",
         "--seq-len", "128"],
        cwd=REPO, capture_output=True, text=True, timeout=600, env=env,
    )
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-2000:])
    rec = _parse_json_line(r.stdout)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "fsdp2"
    # exactly one JSON line (rank 0 only)
    n_json = sum(1 for ln in r.stdout.splitlines() if ln.strip().startswith("{"))
    assert n_json == 1, r.stdout
