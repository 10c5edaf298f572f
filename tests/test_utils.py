"""auto-restart harness test (reference run_with_auto_restart semantics)."""
import sys
import time

from code_intelligence_amd.utils.auto_restart import AutoRestarter, snapshot


def test_restarts_on_file_change(tmp_path):
    watched = tmp_path / "code.py"
    watched.write_text("x = 1\n")
    marker = tmp_path / "marker"
    cmd = [sys.executable, "-c",
           f"import time; open(r'{marker}', 'a').write('run\\n'); time.sleep(30)"]
    r = AutoRestarter(cmd, [str(tmp_path)], poll_s=0.2)
    import threading
    t = threading.Thread(target=r.run, kwargs={"max_iterations": 12}, daemon=True)
    t.start()
    time.sleep(0.8)
    watched.write_text("x = 2\n")  # trigger change
    t.join(timeout=10)
    if r.proc and r.proc.poll() is None:
        r.proc.kill()
    assert r.restarts >= 1
    assert len(marker.read_text().splitlines()) >= 2


def test_snapshot_filters_extensions(tmp_path):
    (tmp_path / "a.py").write_text("x")
    (tmp_path / "b.log").write_text("x")
    snap = snapshot([str(tmp_path)])
    assert any(k.endswith("a.py") for k in snap)
    assert not any(k.endswith("b.log") for k in snap)
