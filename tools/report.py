#!/usr/bin/env python3
"""Render an HTML report from the committed measurement artifacts
(reference analog: test/stress/report.go's HTML output).

    python tools/report.py [--profiles profiles/] [--out report.html]
"""
import argparse
import glob
import html
import json
import os
import sys


def load_json(path):
    try:
        with open(path) as f:
            return json.load(f)
    except Exception:
        return None


def row(cells, tag="td"):
    inner = "".join(f"<{tag}>{html.escape(str(c))}</{tag}>" for c in cells)
    return f"<tr>{inner}</tr>"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--profiles", default=os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "profiles"))
    ap.add_argument("--out", default="report.html")
    args = ap.parse_args()

    sections = []

    # serving benchmarks
    rows = []
    for f in sorted(glob.glob(os.path.join(args.profiles,
                                           "bench_serving_*.json"))):
        d = load_json(f)
        if not d or "total_token_throughput_tok_s" not in d:
            continue
        cfg = d.get("config", {})
        rows.append(row([os.path.basename(f),
                         cfg.get("model", "?"),
                         f"{cfg.get('rate', '?')} req/s",
                         f"{d['total_token_throughput_tok_s']:.0f}",
                         f"{d.get('mean_ttft_ms', 0):.1f}",
                         f"{d.get('mean_itl_ms', 0):.2f}",
                         d.get("completed", "?")]))
    if rows:
        sections.append(
            "<h2>Serving (request-rate)</h2><table>" +
            row(["artifact", "model", "rate", "total tok/s", "mean TTFT ms",
                 "mean ITL ms", "completed"], "th") + "".join(rows) +
            "</table>")

    # offline bench lines
    rows = []
    for f in sorted(glob.glob(os.path.join(args.profiles, "bench_b*.txt"))):
        for line in open(f):
            line = line.strip()
            if not line.startswith("{"):
                continue
            d = load_json_str(line)
            if not d:
                continue
            rows.append(row([os.path.basename(f), d.get("value"),
                             d.get("ms_per_step"),
                             d.get("config", {}).get("global_batch")]))
    if rows:
        sections.append(
            "<h2>Offline bench (bench.py)</h2><table>" +
            row(["artifact", "output tok/s", "ms/step", "batch"], "th") +
            "".join(rows) + "</table>")

    # stress
    d = load_json(os.path.join(args.profiles,
                               "stress_controller_10g_5qps.json"))
    if d:
        rows = [row([phase, s.get("submitted"), s.get("settled"),
                     s.get("p50_s"), s.get("p90_s"), s.get("p99_s")])
                for phase, s in d.items()
                if isinstance(s, dict) and "p50_s" in s]
        sections.append(
            "<h2>Controller stress (real engine processes)</h2><table>" +
            row(["phase", "submitted", "settled", "p50 s", "p90 s",
                 "p99 s"], "th") + "".join(rows) + "</table>")

    # P/D serving + failover (round 2)
    rows = []
    for name, label in (
            ("bench_serving_pd_llama8b_rate10b.json", "P/D disagg 300@10"),
            ("bench_serving_llama8b_rate10_r2.json", "colocated 300@10"),
            ("bench_serving_qwen3_32b_rate10_r2.json", "qwen3-32b 300@10"),
            ("bench_decode_pool_failover.json",
             "decode pool 4x + mid-stream kill")):
        d = load_json(os.path.join(args.profiles, name))
        if d:
            fo = d.get("failover") or {}
            rows.append(row([label, d.get("completed"), d.get("errors"),
                             d.get("total_token_throughput_tok_s"),
                             d.get("mean_ttft_ms"), d.get("mean_itl_ms"),
                             fo.get("recovery_s", "-")]))
    if rows:
        sections.append(
            "<h2>Serving (round 2)</h2><table>" +
            row(["config", "completed", "errors", "tok/s", "mean TTFT ms",
                 "mean ITL ms", "recover s"], "th") +
            "".join(rows) + "</table>")

    # stress matrix (round 2: scenario sweep with controller CPU/RSS)
    d = load_json(os.path.join(args.profiles, "stress_matrix_r2.json"))
    if d and d.get("matrix"):
        rows = []
        for sc in d["matrix"]:
            for phase in ("create", "update", "delete"):
                s = sc.get(phase) or {}
                c = s.get("controller") or {}
                rows.append(row([f"{sc['groups']}g@{sc['qps']}qps", phase,
                                 s.get("p50_s"), s.get("p99_s"),
                                 s.get("timed_out"), c.get("cpu_pct"),
                                 c.get("rss_peak_mb")]))
        sections.append(
            "<h2>Controller stress matrix</h2><table>" +
            row(["scenario", "phase", "p50 s", "p99 s", "timed out",
                 "controller cpu %", "rss MB"], "th") +
            "".join(rows) + "</table>")

    # recovery
    d = load_json(os.path.join(args.profiles, "recovery_llama8b_gpu.json"))
    if d:
        sections.append(
            "<h2>Group recovery</h2><p>model {model} on {device}: detect "
            "{detect_s}s, recover {recovery_s}s (cold start "
            "{cold_start_s}s)</p>".format(**d))

    doc = ("<html><head><meta charset='utf-8'><title>rbg-mi355x report"
           "</title><style>body{font-family:sans-serif;margin:2em}"
           "table{border-collapse:collapse}td,th{border:1px solid #999;"
           "padding:4px 8px}th{background:#eee}</style></head><body>"
           "<h1>rbg-mi355x — measured results</h1>" +
           "".join(sections) + "</body></html>")
    with open(args.out, "w") as f:
        f.write(doc)
    print(f"wrote {args.out} ({len(sections)} sections)")
    return 0


def load_json_str(s):
    try:
        return json.loads(s)
    except Exception:
        return None


if __name__ == "__main__":
    sys.exit(main())
