"""Watchdog + Heartbeat (utils/watchdog.py): hang and stall detection."""
import time

from es_pytorch_amd.utils.watchdog import Heartbeat, Watchdog


def test_watchdog_fires_on_timeout():
    fired = []
    wd = Watchdog(0.2, on_timeout=lambda label: fired.append(label))
    wd.arm("gen 3")
    time.sleep(0.8)
    wd.close()
    assert fired == ["gen 3"]


def test_watchdog_disarm_prevents_firing():
    fired = []
    wd = Watchdog(0.3, on_timeout=lambda label: fired.append(label))
    with wd.guard("fast gen"):
        time.sleep(0.05)
    time.sleep(0.7)
    wd.close()
    assert fired == []


def test_heartbeat_stall_detection(tmp_path):
    folder = str(tmp_path / "hb")
    for rank in (0, 1):
        Heartbeat(folder, rank).beat(gen=5)
    assert len(Heartbeat.read(folder)) == 2
    assert Heartbeat.stalled_ranks(folder, timeout_s=60) == []
    # age rank 1's beat artificially
    import json, os
    p = tmp_path / "hb" / "rank1.json"
    b = json.loads(p.read_text())
    b["ts"] -= 1000
    p.write_text(json.dumps(b))
    stalled = Heartbeat.stalled_ranks(folder, timeout_s=60)
    assert [s["rank"] for s in stalled] == [1] and stalled[0]["gen"] == 5


def test_watchdog_default_action_exits_124():
    """A hung generation becomes a loud exit-124 with stack dumps."""
    import os
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = ("from es_pytorch_amd.utils.watchdog import Watchdog\n"
            "import time\n"
            "wd = Watchdog(0.2)\n"
            "wd.arm('hung gen')\n"
            "time.sleep(30)\n")
    r = subprocess.run([sys.executable, "-c", code], capture_output=True, text=True,
                       timeout=25, env=dict(os.environ, PYTHONPATH=root))
    assert r.returncode == 124
    assert "watchdog" in r.stderr and "hung gen" in r.stderr
    assert "Current thread" in r.stderr or "Thread" in r.stderr  # stack dump


def test_heartbeat_cli(tmp_path, capsys):
    from es_pytorch_amd.utils.watchdog import _main
    Heartbeat(str(tmp_path), 0).beat(gen=12)
    assert _main([str(tmp_path), "--timeout", "300"]) == 0
    assert "rank 0: gen 12" in capsys.readouterr().out
    assert _main([str(tmp_path), "--timeout", "0"]) == 1  # everything stalled
