#!/usr/bin/env python3
"""Micro-benchmarks mirroring the reference's criterion suite
(rust/benches/): message serialize/parse and model<->fixed-point conversion
at 4 B / 100 kB / 1 MB — CPU-side protocol costs that gate coordinator
ingest. Run: python scripts/microbench.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from xaynet_amd import _core

mk = _core.mask
msgmod = _core.message


def timeit(fn, reps=None, min_s=0.3):
    fn()  # warm
    n, t0 = 0, time.perf_counter()
    while True:
        fn()
        n += 1
        dt = time.perf_counter() - t0
        if (reps and n >= reps) or (not reps and dt > min_s):
            return dt / n


def main():
    seed, cpk = b"\x01" * 32, b"\x02" * 32

    # --- sum message to/from bytes (reference benches/messages/sum.rs) ---
    payload = b"\x33" * 64 + b"\x44" * 32
    t = timeit(lambda: msgmod.encode(msgmod.TAG_SUM, payload, seed, cpk))
    wire = msgmod.encode(msgmod.TAG_SUM, payload, seed, cpk)[0]
    print(f"sum_message encode+sign:        {t*1e6:8.2f} us")
    t = timeit(lambda: msgmod.parse_header(wire))
    print(f"sum_message parse header:       {t*1e9:8.1f} ns")
    t = timeit(lambda: msgmod.verify(wire))
    print(f"sum_message parse+verify sig:   {t*1e6:8.2f} us")

    # --- model conversions (reference benches/models/{from,to}_primitives.rs) ---
    cfg = mk.MaskConfig(1, 0, 0, 6)
    pair = mk.MaskConfigPair(cfg, cfg)
    for label, n in (("4B (1 w)", 1), ("100kB (25k w)", 25_000), ("1MB (250k w)", 250_000)):
        w = np.random.default_rng(1).uniform(-1, 1, n).astype(np.float32)
        msk = None

        def mask_once():
            nonlocal msk
            msk = mk.mask_model(b"\x07" * 32, mk.Scalar(1, 2), w, pair)

        t = timeit(mask_once)
        print(f"mask_model f32 {label:14s}: {t*1e3:8.3f} ms  "
              f"({n / t / 1e6:6.1f} Mweights/s)")
        wire_obj = msk.serialize()
        t = timeit(lambda: msk.serialize())
        print(f"  serialize MaskObject:         {t*1e3:8.3f} ms "
              f"({len(wire_obj) / t / 1e6:6.0f} MB/s)")
        t = timeit(lambda: mk.MaskObject.deserialize(bytes(wire_obj)))
        print(f"  deserialize MaskObject:       {t*1e3:8.3f} ms")

    # --- seed -> mask expansion (CPU oracle; the GPU K1 replaces this) ---
    for n in (25_000, 250_000):
        t = timeit(lambda: mk.derive_mask(b"\x05" * 32, n, pair))
        print(f"derive_mask CPU n={n:7d}:      {t*1e3:8.3f} ms "
              f"({n / t / 1e6:6.2f} Mdraws/s)")


def crypto_bench():
    """Ingest-side crypto costs (reference: rayon decrypt pool + Ed25519
    verify; these gate POST /message throughput per worker thread)."""
    cr = _core.crypto
    pk, sk = cr.box_keypair()
    spk, ssk = cr.sign_keypair()
    for label, size in (("sum msg (232 B)", 232), ("update 1k w (7.2 KB)", 7_232),
                        ("update 100k w (700 KB)", 700_232)):
        body = os.urandom(size)
        sealed = cr.sealbox_seal(body, pk)
        t = timeit(lambda: cr.sealbox_open(sealed, pk, sk))
        print(f"sealedbox open {label:22s}: {t*1e6:9.1f} us ({size/t/1e6:7.1f} MB/s)")
    sig = cr.sign_detached(b"x" * 168, ssk)
    t = timeit(lambda: cr.verify_detached(sig, b"x" * 168, spk))
    print(f"ed25519 verify (168 B):          {t*1e6:9.1f} us ({1/t:,.0f}/s per core)")


if __name__ == "__main__":
    crypto_bench()
    main()
