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
