#!/usr/bin/env python
"""Interpret bench.py / SCALE_rNN.json results: weak-scaling efficiency and
a per-rank bubble/communication diagnosis from the embedded stage timers.

Usage:
    python scripts/analyze_scale.py BENCH_r01.json [SCALE_r01.json ...]
    python scripts/analyze_scale.py - < bench_output_line.json
"""

from __future__ import annotations

import json
import sys


def load(path: str):
    text = sys.stdin.read() if path == "-" else open(path).read()
    rows = []
    for line in text.splitlines():
        line = line.strip()
        if not line:
            continue
        try:
            obj = json.loads(line)
        except json.JSONDecodeError:
            continue
        if isinstance(obj, dict) and "value" in obj:
            rows.append(obj)
        elif isinstance(obj, dict):  # driver files may nest results
            for v in obj.values():
                if isinstance(v, dict) and "value" in v:
                    rows.append(v)
    return rows


def main() -> int:
    rows = []
    for p in sys.argv[1:] or ["-"]:
        rows.extend(load(p))
    if not rows:
        print("no bench JSON found")
        return 1
    rows.sort(key=lambda r: r.get("n_gpus", 1))
    base = next((r for r in rows if r.get("n_gpus") == 1), rows[0])
    base_v = base["value"]
    print(f"{'N':>3} {'tok/s':>10} {'ms/step':>9} {'eff':>6}  mbs x gas  ckpt  peakGB")
    for r in rows:
        c = r.get("config", {})
        n = r.get("n_gpus", 1)
        eff = r["value"] / base_v
        print(f"{n:>3} {r['value']:>10.0f} {r['ms_per_step']:>9.0f} {eff:>6.2f}  "
              f"{c.get('micro_batch_size', '?')} x {c.get('grad_accum_steps', '?'):<4} "
              f"{c.get('ckpt_layers_per_stage', '?'):>4}  {c.get('peak_mem_gb', '?')}")
        timers = c.get("stage_timers") or []
        if len(timers) > 1:
            # stage_timers are DEVICE-measured (hipEvent pairs on the compute
            # stream, engine.device_timers): fwd/bwd = kernel time, p2p /
            # allreduce = stalls the compute stream suffered waiting on the
            # comm channels.  p2p on rank 0/last ~= fill-drain bubble; on
            # middle ranks ~= upstream stall.
            steps = r.get("steps", 1)
            worst = max(timers, key=lambda t: t["p2p_s"])
            total = r["ms_per_step"] * steps / 1000.0
            print(f"     per-rank p2p stall (device-true): max {worst['p2p_s']:.1f}s "
                  f"(rank {worst['rank']}, {100 * worst['p2p_s'] / max(total, 1e-9):.0f}% "
                  f"of the timed window)")
            busiest = max(timers, key=lambda t: t["fwd_s"] + t["bwd_s"])
            idle = total - (busiest["fwd_s"] + busiest["bwd_s"] +
                            busiest["p2p_s"] + busiest["allreduce_s"] +
                            busiest["optim_s"])
            print(f"     busiest rank {busiest['rank']}: compute "
                  f"{busiest['fwd_s'] + busiest['bwd_s']:.1f}s, unattributed "
                  f"{idle:.1f}s ({100 * idle / max(total, 1e-9):.0f}% — host gaps "
                  f"/ dataloader if large)")
            ar = max(t["allreduce_s"] for t in timers)
            opt = max(t["optim_s"] for t in timers)
            print(f"     allreduce stall max {ar:.1f}s; optimizer max {opt:.1f}s")
    mbs_note = {r.get('n_gpus'): r.get('config', {}).get('micro_batch_size') for r in rows}
    if len(set(mbs_note.values())) > 1:
        print("note: microbatch size varies by N (adaptive policy) — efficiency "
              "mixes bubble AND kernel-efficiency effects; see "
              "lpp_amd/utils/schedule.py and BACKLOG.md item 1.")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
