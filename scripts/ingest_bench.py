#!/usr/bin/env python3
"""Ingest-path benchmark: decrypt + validate + stage + H2D + aggregate, end
to end through the production serve plane.

Forges one PET round with `--clients` real update messages (signed, eligible,
optionally sealed-box encrypted) against a STAGED coordinator driven by the
multi-process serve plane, then times:
  - the injection window (REST-worker analog: T threads calling
    handle_encrypted_message / handle_message_bytes, GIL released)
  - the Update phase wall (first accept -> phase close)
  - ingest-to-model wall (first accept -> global model published)

Usage (GPU box):
  python scripts/ingest_bench.py --length 1000000 --clients 512 --workers 1
  python scripts/ingest_bench.py --length 25000000 --clients 64 --plain
"""
import argparse
import faulthandler
import json
import os
import sys
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from xaynet_amd import _core  # noqa: E402

co = _core.coordinator
mk = _core.mask
cr = _core.crypto
msgmod = _core.message
sdk = _core.sdk


def eligible_seed(rng, seed_round, want_sum, sum_prob, update_prob):
    for _ in range(4000):
        sgn = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        pk, sk = cr.sign_keypair_from_seed(sgn)
        sum_sig = cr.sign_detached(seed_round + b"sum", sk)
        upd_sig = cr.sign_detached(seed_round + b"update", sk)
        is_sum = cr.is_eligible(sum_sig, sum_prob)
        if want_sum and is_sum:
            return sgn, sk, sum_sig, upd_sig
        if not want_sum and not is_sum and cr.is_eligible(upd_sig, update_prob):
            return sgn, sk, sum_sig, upd_sig
    raise AssertionError("no eligible seed found")


def pack_values(vals: np.ndarray, bpn: int) -> bytes:
    out = np.zeros((len(vals), bpn), dtype=np.uint8)
    v = vals.astype(np.uint64)
    for b in range(bpn):
        out[:, b] = ((v >> np.uint64(8 * b)) & np.uint64(0xFF)).astype(np.uint8)
    return out.tobytes()


def main():
    if os.environ.get("INGEST_DEBUG"):
        faulthandler.dump_traceback_later(60, exit=True)
    ap = argparse.ArgumentParser()
    ap.add_argument("--length", type=int, default=1_000_000)
    ap.add_argument("--clients", type=int, default=512)
    ap.add_argument("--workers", type=int, default=1)
    ap.add_argument("--device", choices=["cuda", "cpu"], default="cuda")
    ap.add_argument("--threads", type=int, default=32, help="injector threads")
    ap.add_argument("--plain", action="store_true",
                    help="skip sealbox (measures post-decrypt pipeline only)")
    ap.add_argument("--slots", type=int, default=32)
    args = ap.parse_args()

    sum_prob, update_prob = 0.5, 0.999
    s = co.Settings()
    s.sum_prob = sum_prob
    s.update_prob = update_prob
    s.model_length = args.length
    c = mk.MaskConfig(1, 0, 0, 6)  # Prime/F32/B0/M6, bpn=7
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 1, 0.05, 30.0)
    # count.min = clients: the gate closes at count.min once time.min has
    # elapsed (reference handler.rs), and we want the whole batch timed
    s.set_update(args.clients, args.clients, 0.2, 600.0)
    s.set_sum2(1, 1, 0.05, 60.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), True)

    from xaynet_amd.parallel.serve import MultiGpuServeDriver

    driver = MultiGpuServeDriver(coord, c, c, args.length, n_workers=args.workers,
                                 device_kind=args.device, slots_per_worker=args.slots)
    driver.start()

    rng = np.random.default_rng(11)
    order = int(c.order)
    bpn = c.bytes_per_number
    E = co.PipelineError

    coord.run_one_phase()  # Idle -> Sum
    params = bytes(coord.fetch_round_params())
    cpk, seed_round = params[:32], params[48:80]

    # --- sum participant ---
    s_sgn, s_sk, s_sum_sig, _ = eligible_seed(rng, seed_round, True, sum_prob, update_prob)
    ephm_pk, ephm_sk = cr.box_keypair()
    sum_wire = bytes(msgmod.encode(msgmod.TAG_SUM, bytes(s_sum_sig) + ephm_pk, s_sgn, cpk)[0])
    t = threading.Thread(target=coord.run_one_phase, daemon=True)
    t.start()
    time.sleep(0.02)
    assert coord.handle_message_bytes(sum_wire) == int(E.Ok)
    t.join(60)
    assert coord.phase == co.PhaseId.Update, coord.phase
    sum_pk = cr.sign_keypair_from_seed(s_sgn)[0]

    # --- forge updates (untimed). All clients share one mask seed/payload;
    # the aggregated mask is then clients * mask(seed) mod order. ---
    t_forge = time.time()
    mask_seed = b"\x02" * 32
    weights = rng.uniform(-1, 1, args.length).astype(np.float64)
    masked = mk.mask_model(mask_seed, mk.Scalar(1, args.clients), weights,
                           mk.MaskConfigPair(c, c))
    masked_wire = bytes(masked.serialize())
    sealed_seed = cr.sealbox_seal(mask_seed, ephm_pk)
    entry = bytes(sum_pk) + bytes(sealed_seed)
    seed_dict_body = (4 + len(entry)).to_bytes(4, "big") + entry  # inclusive len

    def forge(i):
        sgn, sk, sum_sig, upd_sig = eligible_seed(
            np.random.default_rng(1000 + i), seed_round, False, sum_prob, update_prob)
        payload = bytes(sum_sig) + bytes(upd_sig) + masked_wire + seed_dict_body
        wire = bytes(msgmod.encode(msgmod.TAG_UPDATE, payload, sgn, cpk,
                                   max_payload=1 << 62)[0])
        if args.plain:
            return wire
        return bytes(cr.sealbox_seal(wire, cpk))

    with ThreadPoolExecutor(max_workers=args.threads) as ex:
        messages = list(ex.map(forge, range(args.clients)))
    forge_s = time.time() - t_forge
    total_mb = sum(len(m) for m in messages) / 1e6
    print(f"forged {args.clients} update messages ({total_mb:.0f} MB) in {forge_s:.1f}s",
          file=sys.stderr)

    print("deriving aggregated mask...", file=sys.stderr, flush=True)
    # --- aggregated mask for sum2 (untimed) ---
    mobj = mk.derive_mask(mask_seed, args.length, mk.MaskConfigPair(c, c))
    mvals = np.frombuffer(mobj.vect_bytes, dtype=np.uint8).reshape(args.length, bpn)
    v = np.zeros(args.length, dtype=np.uint64)
    for b in range(bpn):
        v |= mvals[:, b].astype(np.uint64) << np.uint64(8 * b)
    agg_vals = ((v.astype(object) * args.clients) % order).astype(np.uint64)
    unit_mask = (int(mobj.unit_value) * args.clients) % order
    agg_mask_wire = (
        masked_wire[:4] + args.length.to_bytes(4, "big") + pack_values(agg_vals, bpn)
        + masked_wire[:4] + unit_mask.to_bytes(bpn, "little")
    )
    s2_payload = bytes(s_sum_sig) + agg_mask_wire
    sum2_wire = bytes(msgmod.encode(msgmod.TAG_SUM2, s2_payload, s_sgn, cpk,
                                    max_payload=1 << 62)[0])

    print("injecting...", file=sys.stderr, flush=True)
    # --- timed injection ---
    inject = (coord.handle_message_bytes if args.plain else coord.handle_encrypted_message)
    results = []
    t_upd = threading.Thread(target=coord.run_one_phase, daemon=True)
    t_upd.start()
    time.sleep(0.02)
    t0 = time.time()
    with ThreadPoolExecutor(max_workers=args.threads) as ex:
        results = list(ex.map(inject, messages))
    t1 = time.time()
    accepted = sum(1 for r in results if r == int(E.Ok))
    print(f"injected in {t1-t0:.2f}s, accepted {accepted}", file=sys.stderr, flush=True)
    t_upd.join(600)
    print("update phase closed", file=sys.stderr, flush=True)
    t2 = time.time()
    assert coord.phase == co.PhaseId.Sum2, coord.phase

    # --- sum2 + unmask (driver supplies the model) ---
    t_s2 = threading.Thread(target=coord.run_one_phase, daemon=True)
    t_s2.start()
    time.sleep(0.02)
    assert coord.handle_message_bytes(sum2_wire) == int(E.Ok)
    t_s2.join(120)
    print("sum2 done; unmasking...", file=sys.stderr, flush=True)
    coord.run_one_phase()  # Unmask (blocks on driver)
    t3 = time.time()
    body = coord.fetch_model()
    assert body and body[0] == 1, "no global model"
    model = np.asarray(sdk.decode_model(body, 0))
    err = np.abs(model - weights.astype(np.float32)).max()

    driver.stop()
    coord.stop()

    inj = t1 - t0
    res = {
        "metric": "ingest updates/s (decrypt+validate+stage+H2D+aggregate, serve plane)",
        "length": args.length,
        "clients": args.clients,
        "accepted": accepted,
        "workers": args.workers,
        "device": args.device,
        "sealed": not args.plain,
        "inject_threads": args.threads,
        "inject_window_s": round(inj, 3),
        "inject_updates_per_s": round(accepted / inj, 1),
        "inject_MB_per_s": round(total_mb / inj, 1),
        "update_phase_s": round(t2 - t0, 3),
        "ingest_to_model_s": round(t3 - t0, 3),
        "e2e_updates_per_s": round(accepted / (t3 - t0), 1),
        "model_max_err": float(err),
    }
    print(json.dumps(res))
    assert accepted == args.clients, f"only {accepted}/{args.clients} accepted"
    assert err < 2e-4, f"model mismatch {err}"


if __name__ == "__main__":
    main()
