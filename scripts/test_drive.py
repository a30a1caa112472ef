#!/usr/bin/env python3
"""test-drive: N participants against a coordinator over HTTP — the
reference's rust/examples/test-drive scale harness equivalent.

Self-hosted mode spins up the native coordinator + REST server in-process:
  python scripts/test_drive.py --serve -n 50 --rounds 3 --length 1000
Against an external coordinator (e.g. `python -m xaynet_amd.server`):
  python scripts/test_drive.py --url http://127.0.0.1:8081 -n 50
"""
import argparse
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from xaynet_amd import _core

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask
rest = _core.rest


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--url", default=None, help="coordinator url (default: self-serve)")
    ap.add_argument("--serve", action="store_true", help="run a coordinator in-process")
    ap.add_argument("-n", "--participants", type=int, default=20)
    ap.add_argument("--rounds", type=int, default=2)
    ap.add_argument("--length", type=int, default=1000)
    ap.add_argument("--update-min", type=int, default=3)
    ap.add_argument("--sum-prob", type=float, default=0.3)
    ap.add_argument("--timeout", type=float, default=120.0)
    ap.add_argument("--gpu", action="store_true", help="staged GPU aggregation plane")
    ap.add_argument("--workers", type=int, default=1,
                    help="serve-plane worker processes (one per GPU; >1 = the "
                         "multi-GPU sharded plane, also runnable on CPU for CI)")
    ap.add_argument("--gpu-clients", action="store_true",
                    help="attach the MI355X participant accelerator (K1+K5w "
                         "update masking, K1+K2 sum2 aggregation)")
    ap.add_argument("--worker-device", choices=["cuda", "cpu"], default=None,
                    help="serve-plane device kind (default: cuda with --gpu, "
                         "cpu otherwise)")
    ap.add_argument("--max-message-size", type=int, default=0,
                    help="SDK chunking threshold (0 = reference default 4096-184)")
    ap.add_argument("--mask-config", default="f32-m6",
                    choices=["f32-m6", "i64-m6", "f32-m3", "f64-m3", "f64-b4-m9"],
                    help="mask config preset (f64-m3 has a >64-bit group order "
                    "— exercises the wide GPU path)")
    ap.add_argument("--metrics", default=None,
                    help="write coordinator metrics (InfluxDB line protocol) to this file")
    args = ap.parse_args()

    server = coord = driver = None
    if args.metrics:
        co.install_metrics_file(args.metrics)
    if args.url is None or args.serve:
        s = co.Settings()
        s.sum_prob = args.sum_prob
        s.update_prob = 0.999
        s.model_length = args.length
        c = {  # group, data_type, bound_type, model_type
            "f32-m6": mk.MaskConfig(1, 0, 0, 6),  # Prime/F32/B0/M6
            "i64-m6": mk.MaskConfig(1, 3, 0, 6),  # Prime/I64/B0/M6
            "f32-m3": mk.MaskConfig(1, 0, 0, 3),  # Prime/F32/B0/M3
            "f64-m3": mk.MaskConfig(1, 1, 0, 3),  # Prime/F64/B0/M3 (bpn=10, wide)
            "f64-b4-m9": mk.MaskConfig(1, 1, 2, 9),  # Prime/F64/B4/M9 (bpn=13/14, wide)
        }[args.mask_config]
        s.mask_cfg = mk.MaskConfigPair(c, c)
        s.set_sum(1, max(10, args.participants), 0.2, 30.0)
        s.set_update(args.update_min, args.participants, 0.2, 30.0)
        s.set_sum2(1, max(10, args.participants), 0.2, 30.0)
        staged = args.gpu or args.workers > 1 or args.worker_device is not None
        coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), staged)
        server = rest.RestServer(coord, "127.0.0.1", 0, 8)
        assert server.start()
        if args.workers > 1 or args.worker_device is not None:
            from xaynet_amd.parallel.serve import MultiGpuServeDriver

            kind = args.worker_device or ("cuda" if args.gpu else "cpu")
            driver = MultiGpuServeDriver(coord, c, c, args.length,
                                         n_workers=args.workers, device_kind=kind)
            driver.start()
        elif args.gpu:
            from xaynet_amd.ops import make_coordinator_driver

            driver = make_coordinator_driver(coord, c, c, args.length)
            driver.start()
        coord.start()
        url = f"http://127.0.0.1:{server.port}"
    else:
        url = args.url

    host, port = url.split("//")[-1].rsplit(":", 1)
    rng = np.random.default_rng(1234)
    dtype = {"f32-m6": np.float32, "i64-m6": np.int64,
             "f32-m3": np.float32, "f64-m3": np.float64,
             "f64-b4-m9": np.float64}[args.mask_config]
    if dtype is np.int64:
        weights = rng.integers(-1000, 1000, args.length).astype(np.int64)
    else:
        weights = rng.uniform(-1, 1, args.length).astype(dtype)

    accel = None
    if args.gpu_clients:
        from xaynet_amd.ops.accel import ParticipantAccel

        accel = ParticipantAccel()  # auto-configures per round

    stop = threading.Event()
    # per-participant count of new-global-model events; a round is complete
    # when any participant's count advances (models may be byte-identical
    # between rounds, so counting distinct bodies would undercount)
    round_events = []
    lock = threading.Lock()

    def agent(i):
        client = rest.HttpXaynetClient(host, int(port))
        p = sdk.Participant(
            bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client,
            max_message_size=args.max_message_size,
        )
        if accel is not None:
            accel.attach(p)
        seen = 0
        while not stop.is_set():
            p.tick()
            if p.should_set_model:
                p.set_model(weights)
            if p.new_global_model:
                body = p.global_model_bincode()  # consumes the flag
                if body is not None:
                    seen += 1
                    with lock:
                        round_events.append((i, seen, time.time()))
            time.sleep(0.01 if p.made_progress else 0.05)

    threads = [threading.Thread(target=agent, args=(i,), daemon=True) for i in range(args.participants)]
    t0 = time.time()
    for t in threads:
        t.start()

    rounds_done = 0
    round_times = []
    try:
        while time.time() - t0 < args.timeout and rounds_done < args.rounds:
            with lock:
                counts = {}
                for i, seen, ts in round_events:
                    counts[seen] = min(counts.get(seen, ts), ts)
                rounds_done = max(counts) if counts else 0
                round_times = [counts[k] for k in sorted(counts)]
            time.sleep(0.1)
    finally:
        stop.set()
        for t in threads:
            t.join(timeout=2)
        if coord:
            coord.stop()
        if driver:
            driver.stop()
        if server:
            server.stop()

    elapsed = time.time() - t0
    ok = rounds_done >= args.rounds
    print(
        f"test-drive: {args.participants} participants, {rounds_done} rounds "
        f"in {elapsed:.1f}s ({'OK' if ok else 'TIMEOUT'})"
    )
    if len(round_times) >= 2:
        gaps = [b - a for a, b in zip(round_times, round_times[1:])]
        print(f"round wall-clock: min {min(gaps):.2f}s avg {sum(gaps)/len(gaps):.2f}s")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
