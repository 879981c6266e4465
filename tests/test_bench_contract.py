"""bench.py driver-contract test: runs the real script (CPU, tiny) and
validates the JSON line fields the driver parses."""
import json
import subprocess
import sys


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "32", "--vocab", "1000", "--n-cat", "4"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
              "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
              "dtype", "data", "config"):
        assert k in d, f"missing field {k}"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0
    assert "global_batch" in d["config"] and "parallelism" in d["config"]


def test_bench_torchrun_2rank_contract():
    """The driver's multi-GPU launch path: torchrun world=2 (gloo on CPU),
    EP embeddings, JSON from rank 0 only."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29651", "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "32", "--vocab", "1000", "--n-cat", "4"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got {len(lines)}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["global_batch"] == 64
    assert "ep2" in d["config"]["parallelism"]


def test_bench_torchrun_2rank_dp_mode():
    """--emb-mode dp at world=2: replicated arenas + sparse allgather
    aggregation through the real bench script."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29653", "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "32", "--vocab", "1000", "--n-cat", "4",
         "--emb-mode", "dp"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    d = json.loads(lines[0])
    assert d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0


def test_bench_torchrun_4rank_table_ep():
    """world=4 (gloo CPU): the table-EP feature assignment puts 1 feature on
    each of 4 ranks for n-cat=4 — a different split shape than the 2-rank
    tests; exactly what the driver's SCALE run exercises at N=4."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29655", "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "16", "--vocab", "500", "--n-cat", "6"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    d = json.loads(lines[0])
    assert d["n_gpus"] == 4
    assert "ep4" in d["config"]["parallelism"]


def test_bench_torchrun_8rank_table_ep():
    """world=8 (gloo CPU): the driver's SCALE shape at N=8.  With n-cat=26
    (the headline feature count) the greedy assignment puts 3-4 tables on
    each rank; checks the full 8-way static all-to-all path end to end."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29657", "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "16", "--vocab", "200", "--n-cat", "26"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    d = json.loads(lines[0])
    assert d["n_gpus"] == 8
    assert d["config"]["global_batch"] == 128
    assert "ep8" in d["config"]["parallelism"]


def test_bench_torchrun_2rank_ep_row():
    """--emb-mode ep_row at world=2: row%world-sharded arenas through the
    real bench script (the beyond-HBM single-table layout)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29659", "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "32", "--vocab", "1000", "--n-cat", "4",
         "--emb-mode", "ep_row"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
