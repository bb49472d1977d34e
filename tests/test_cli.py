"""CLI smoke (R1 entry/config parity): python -m distributedes_amd."""
import subprocess
import sys


def test_cli_runs_cartpole(tmp_path):
    ck = tmp_path / "ck.pt"
    r = subprocess.run(
        [sys.executable, "-m", "distributedes_amd", "--preset", "cartpole_es",
         "--ops", "torch_ref",
         "--set", "pop=8", "generations=2", "episode_steps=10",
         "noise_table_size=65536", "device='cpu'",
         f"checkpoint_path='{ck}'"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert ck.exists()
