"""viz: log parsing and plot generation (reference src/utils/viz.py:28-79).

Exercises the exact round trip the production loop produces: reporter-format
``k:v`` log lines -> parse_log -> graph_log PNG, and DefaultReporterSet's
per-gen fitness .npy dumps -> graph_fits PNG."""
import os

import numpy as np

from es_pytorch_amd.utils.viz import graph_fits, graph_log, parse_log


def _write_log(path, gens=5):
    with open(path, "w") as f:
        for g in range(gens):
            f.write("\n----------------------------------------\n")
            f.write(f"gen:{g}\navg-0:{g * 10.0}\nmax-0:{g * 12.0}\n"
                    f"dist:{g * 0.5}\nrew:{100.0 + g}\n\nsteps:1000\n"
                    f"cum steps:{1000 * (g + 1)}\nn fits ranked:64\ntime:0.06\n")


def test_parse_log_reporter_format(tmp_path):
    p = tmp_path / "es.log"
    _write_log(p)
    s = parse_log(str(p))
    assert s["rew"] == [100.0, 101.0, 102.0, 103.0, 104.0]
    assert s["cum steps"][-1] == 5000.0
    assert len(s["time"]) == 5


def test_parse_log_logging_prefix(tmp_path):
    """LoggerReporter lines carry the INFO:root: prefix."""
    p = tmp_path / "es.log"
    with open(p, "w") as f:
        f.write("INFO:root:rew:42.5\nINFO:root:dist:1.25\n")
    s = parse_log(str(p))
    assert s["rew"] == [42.5] and s["dist"] == [1.25]


def test_graph_log_writes_png(tmp_path):
    p = tmp_path / "es.log"
    _write_log(p)
    out = tmp_path / "curve.png"
    graph_log(str(p), out=str(out))
    assert os.path.getsize(out) > 1000  # a real PNG, not an empty touch


def test_graph_fits_reads_reporter_dumps(tmp_path):
    fits_dir = tmp_path / "fits"
    os.makedirs(fits_dir)
    rng = np.random.RandomState(0)
    for g in range(4):
        np.save(str(fits_dir / f"{g}.np"), rng.randn(16, 1) + g)
    out = tmp_path / "fits.png"
    graph_fits(str(fits_dir), out=str(out))
    assert os.path.getsize(out) > 1000
