"""Rot guards for the stress harness and HTML report generator
(reference test/stress/* + report.go analogs): a tiny end-to-end run of
each so CLI drift or schema drift in profiles/ breaks the suite, not the
next stress campaign."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_stress_harness_minimal_run(tmp_path):
    out = tmp_path / "stress.json"
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "stress.py"),
         "--groups", "2", "--qps", "20", "--roles", "1",
         "--replicas", "1", "--timeout", "60", "--out", str(out)],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-2000:]
    doc = json.loads(out.read_text())
    # the report schema the HTML generator and BENCHMARKS.md consume
    assert doc["config"]["groups"] == 2
    for phase in ("create", "update", "delete"):
        ph = doc[phase]
        assert {"p50_s", "p90_s", "p99_s", "achieved_qps"} <= set(ph)
        assert ph["submitted"] == 2 and ph["settled"] == 2
        assert ph["timed_out"] == 0


def test_report_generator_writes_html(tmp_path):
    out = tmp_path / "report.html"
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "report.py"),
         "--out", str(out)],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-2000:]
    html = out.read_text()
    assert html.lstrip().lower().startswith("<!doctype html") or "<html" in html
    assert "stress" in html.lower() or "bench" in html.lower()
