"""PhaseTimer + CLI smoke tests."""
import json
import subprocess
import sys
import time

from hefl.utils import PhaseTimer, format_phase_table


def test_phase_timer_cpu():
    t = PhaseTimer(use_gpu_events=False)
    with t.phase("a"):
        time.sleep(0.01)
    with t.phase("a"):
        pass
    with t.phase("b"):
        pass
    s = t.summary()
    assert s["a"]["calls"] == 2 and s["a"]["wall_s"] >= 0.01
    assert "phase" in format_phase_table(s)


def test_cli_runs_tiny_experiment():
    out = subprocess.run(
        [sys.executable, "-m", "hefl", "--preset", "config1", "--rounds", "1",
         "--epochs", "1", "--clients", "2", "--device", "cpu", "--json"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    last = out.stdout.strip().splitlines()[-1]
    d = json.loads(last)
    assert d["clients"] == 2 and d["encrypted"] is False
    assert 0.0 <= d["metrics"]["accuracy"] <= 1.0


def test_bench_json_contract(tmp_path):
    """bench.py is the driver's measurement contract: run it tiny on CPU and
    validate the single JSON line it prints (keys, metric name, weak
    scaling, vs_baseline arithmetic)."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.join(os.path.dirname(__file__), "..")
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--local-epochs", "1"],
        cwd=repo, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"] == "fl_rounds_per_sec"
    assert d["unit"] == "rounds/s" and d["higher_is_better"] is True
    assert d["scaling"] == "weak" and d["n_gpus"] == 1
    assert d["steps"] == 1 and d["warmup"] == 0
    assert d["data"] == "synthetic"
    assert abs(d["vs_baseline"] - d["value"] / 1.52e-4) < 1e-6
    assert abs(d["ms_per_step"] - 1000.0 / d["value"]) < 1e-6
    cfg = d["config"]
    assert cfg["model"] == "cnn2" and cfg["samples_per_client"] == 720
    assert cfg["he"]["scheme"] == "CKKS" and cfg["he"]["encrypted"] is True


def test_bench_two_rank_driver_invocation():
    """The driver launches bench.py via torch.distributed.run for the
    scaling bench; exercise that exact invocation at world=2 on CPU/gloo
    and check the rank-0 JSON (weak scaling: n_gpus=2, global batch 2x)."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.join(os.path.dirname(__file__), "..")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(__import__("conftest").free_port()), "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--local-epochs", "1"],
        cwd=repo, capture_output=True, text=True, timeout=600,
        env=dict(os.environ, MASTER_ADDR="127.0.0.1"))
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2 and d["scaling"] == "weak"
    assert d["config"]["global_batch"] == 64
    assert d["config"]["parallelism"].startswith("fl-dp2")


def test_cli_checkpoint_resume(tmp_path):
    """--checkpoint saves per-round state and resumes past completed rounds
    (the round-granularity recovery story)."""
    import subprocess
    import sys
    ck = str(tmp_path / "round.pt")
    base = [sys.executable, "-m", "hefl", "--preset", "config1", "--rounds",
            "1", "--epochs", "1", "--clients", "2", "--checkpoint", ck,
            "--json"]
    r1 = subprocess.run(base, capture_output=True, text=True, timeout=240)
    assert r1.returncode == 0, r1.stderr[-800:]
    import os
    assert os.path.exists(ck)
    # second invocation with the same target round count resumes and exits
    r2 = subprocess.run(base, capture_output=True, text=True, timeout=240)
    assert r2.returncode == 0, r2.stderr[-800:]
    assert "resumed from" in r2.stdout and "nothing to do" in r2.stdout
    # raising --rounds continues from the checkpoint
    r3 = subprocess.run(base[:6] + ["2"] + base[7:], capture_output=True,
                        text=True, timeout=240)
    assert r3.returncode == 0, r3.stderr[-800:]
    assert "round 1:" in r3.stdout and "round 0:" not in r3.stdout
