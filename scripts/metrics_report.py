#!/usr/bin/env python3
"""Summarize a coordinator metrics file (InfluxDB line protocol, the
`[metrics.influxdb] url = "file:<path>"` sink) into per-round phase timings
and message counters — the observability view the reference builds in
Grafana on top of InfluxDB.

Usage: python scripts/metrics_report.py /path/to/metrics.lp
"""
import sys
from collections import defaultdict

PHASES = {0: "Idle", 1: "Sum", 2: "Update", 3: "Sum2", 4: "Unmask", 5: "Failure", 6: "Shutdown"}


def parse(path):
    rows = []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            head, _, rest = line.partition(" value=")
            value, _, ts = rest.partition(" ")
            parts = head.split(",")
            tags = dict(p.split("=", 1) for p in parts[1:])
            rows.append((parts[0], tags, float(value), int(ts)))
    return rows


def main(path):
    rows = parse(path)
    # phase transitions -> durations
    transitions = [(int(t["round_id"]), int(v), ts) for m, t, v, ts in rows if m == "phase"]
    counters = defaultdict(int)
    for m, t, v, ts in rows:
        if m.startswith("message_"):
            counters[(int(t["round_id"]), int(t["phase"]), m)] += int(v)

    print(f"{'round':>5} {'phase':>8} {'duration':>10}  messages (acc/rej/disc)")
    for i in range(len(transitions) - 1):
        rid, ph, ts = transitions[i]
        _, _, ts_next = transitions[i + 1]
        dur = (ts_next - ts) / 1e9
        acc = counters.get((rid, ph, "message_accepted"), 0)
        rej = counters.get((rid, ph, "message_rejected"), 0)
        dis = counters.get((rid, ph, "message_discarded"), 0)
        msgs = f"{acc}/{rej}/{dis}" if acc or rej or dis else ""
        print(f"{rid:>5} {PHASES.get(ph, ph):>8} {dur:>9.3f}s  {msgs}")

    rounds = {t[0] for t in transitions}
    masks = {int(t["round_id"]): v for m, t, v, ts in rows if m == "masks_total_number"}
    print(f"\nrounds seen: {len(rounds)}; unmask reached in {len(masks)}; "
          f"total accepted: {sum(v for k, v in counters.items() if k[2] == 'message_accepted')}")


if __name__ == "__main__":
    sys.exit(main(sys.argv[1]) if len(sys.argv) > 1 else print(__doc__))
