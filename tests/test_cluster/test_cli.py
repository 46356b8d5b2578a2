"""CLI surface (reference: colossalai/cli — run / check commands)."""

import subprocess
import sys


def test_check_command():
    r = subprocess.run([sys.executable, "-m", "colossalai_amd.cli.cli", "check"],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    assert "torch:" in r.stdout


def test_run_command_single_proc(tmp_path):
    import colossalai_amd as pkg

    repo = str(__import__("pathlib").Path(pkg.__file__).parent.parent)
    script = tmp_path / "hello.py"
    # torchrun children get the SCRIPT dir as sys.path[0], not the cwd
    script.write_text(f"import sys; sys.path.insert(0, {repo!r})\n"
                      "import os, colossalai_amd\n"
                      "colossalai_amd.launch_from_torch(backend='gloo', verbose=False)\n"
                      "print('rank', os.environ['RANK'], 'ok')\n")
    r = subprocess.run([sys.executable, "-m", "colossalai_amd.cli.cli", "run",
                        "--nproc_per_node", "2", "--master_port", "29771", str(script)],
                       capture_output=True, text=True, timeout=180)
    assert r.returncode == 0, r.stderr[-800:]
    assert "ok" in r.stdout
