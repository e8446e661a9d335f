"""bench.py launch-contract rehearsal (what the driver runs at round end).

The driver launches `python -m torch.distributed.run --nnodes=1
--nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...` — this
test runs that EXACT shape at N=8 on CPU/gloo (tiny model) and checks
the one-line JSON contract, plus the honest-n_gpus guard."""

import json
import subprocess
import sys


def test_torchrun_world8_bench_contract(tmp_path):
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29751", "bench.py", "--gpus", "8",
         "--model", "tiny-qwen3", "--steps", "4", "--warmup", "1",
         "--concurrency", "4", "--prompt-len", "48", "--gen-len", "6"],
        capture_output=True, text=True, timeout=420,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 8
    assert rec["steps"] == 4 and rec["warmup"] == 1
    assert rec["scaling"] == "weak" and rec["dtype"] == "bf16"
    assert rec["config"]["parallelism"] == "dp8"
    assert rec["value"] > 0 and rec["higher_is_better"] is True


def test_bench_refuses_dishonest_gpus():
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "4", "--steps", "1"],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode != 0
    assert "WORLD_SIZE=1" in out.stderr
