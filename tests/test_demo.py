"""The end-to-end demo (examples/end_to_end.py) runs clean on CPU."""
import subprocess
import sys


def test_end_to_end_demo(tmp_path):
    out = subprocess.run(
        [sys.executable, "examples/end_to_end.py", "--epochs", "1",
         "--rows", "400", "--outdir", str(tmp_path)],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-800:]
    assert "predictions" in out.stdout
