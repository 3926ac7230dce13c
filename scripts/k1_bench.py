"""K1 A/B bench: v1 (pairing-per-lane) vs v2 (2-lane Fp2-decomposed).

Measures launch latency and verifies/s at several batch sizes on one
MI355X.  Signatures are generated once on the host (64 distinct keypairs,
replicated to fill big batches — identical per-lane work either way) and
both kernels' verdicts are cross-checked against the host verdict.

Usage (on the GPU box):
    python scripts/k1_bench.py [--batches 1024,4096,10240,65536] [--reps 3]
"""

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from pushcdn_amd.crypto import bls
from pushcdn_amd.ops import get_gpu_ops
from pushcdn_amd.ops.build import build_core


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batches", default="1024,4096,10240,65536")
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--distinct", type=int, default=64)
    args = ap.parse_args()

    ops = get_gpu_ops()
    core = build_core()
    ns = bls.USER_MARSHAL_NAMESPACE

    D = args.distinct
    base = []
    for i in range(D):
        kp = bls.KeyPair.from_seed(i)
        msg = f"k1bench-{i}".encode()
        sig = bls.sign(kp.private_key, ns, msg)
        if i % 5 == 0:  # some invalid lanes so both paths exercise failure
            sig = bytes([sig[0] ^ 1]) + sig[1:]
        want = 1 if core.verify(kp.public_key, ns, msg, sig) else 0
        base.append((kp.public_key, sig, ns.encode() + msg + b"\x00", want))

    probe = torch.zeros(1, dtype=torch.uint8, device="cuda")
    lines = ops.precompute_g2_lines(probe)
    torch.cuda.synchronize()

    results = []
    for N in [int(x) for x in args.batches.split(",")]:
        vks = bytearray()
        sigs = bytearray()
        msgs = bytearray()
        offsets = [0]
        want = []
        for i in range(N):
            pk, sig, m, w = base[i % D]
            vks += pk
            sigs += sig
            msgs += m
            offsets.append(len(msgs))
            want.append(w)
        vks_t = torch.frombuffer(vks, dtype=torch.uint8).to("cuda")
        sigs_t = torch.frombuffer(sigs, dtype=torch.uint8).to("cuda")
        msgs_t = torch.frombuffer(msgs, dtype=torch.uint8).to("cuda")
        moff_t = torch.tensor(offsets, dtype=torch.int64, device="cuda")

        import secrets as _secrets

        rand_r = torch.frombuffer(bytearray(_secrets.token_bytes(8 * N)),
                                  dtype=torch.int64).to("cuda")
        # an ALL-VALID batch for the v3 clean path (auth storms are
        # overwhelmingly valid; the 20%-invalid batch above exercises the
        # per-wave exact fallback instead)
        cvks = bytearray(); csigs = bytearray()
        good = [b for b in base if b[3] == 1]
        for i in range(N):
            pk, sig, _m2, _w = good[i % len(good)]
            cvks += pk; csigs += sig
        cmsgs = bytearray(); coffs = [0]
        for i in range(N):
            _pk, _sig, m2, _w = good[i % len(good)]
            cmsgs += m2; coffs.append(len(cmsgs))
        cvks_t = torch.frombuffer(cvks, dtype=torch.uint8).to("cuda")
        csigs_t = torch.frombuffer(csigs, dtype=torch.uint8).to("cuda")
        cmsgs_t = torch.frombuffer(cmsgs, dtype=torch.uint8).to("cuda")
        cmoff_t = torch.tensor(coffs, dtype=torch.int64, device="cuda")

        row = {"batch": N}
        for name, fn, expect in (
            ("v1", lambda: ops.bls_verify_batch(vks_t, sigs_t, msgs_t, moff_t), want),
            ("v2", lambda: ops.bls_verify_batch2(vks_t, sigs_t, msgs_t, moff_t, lines),
             want),
            ("v3_mixed", lambda: ops.bls_verify_batch_wave(vks_t, sigs_t, msgs_t,
                                                           moff_t, lines, rand_r), want),
            ("v3_clean", lambda: ops.bls_verify_batch_wave(cvks_t, csigs_t, cmsgs_t,
                                                           cmoff_t, lines, rand_r),
             [1] * N),
        ):
            ok = fn()
            torch.cuda.synchronize()
            assert ok.cpu().tolist() == expect, f"{name} wrong verdicts at N={N}"
            ts = []
            for _ in range(args.reps):
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                fn()
                torch.cuda.synchronize()
                ts.append(time.perf_counter() - t0)
            best = min(ts)
            row[name + "_ms"] = round(best * 1e3, 2)
            row[name + "_vps"] = round(N / best)
        row["v2_speedup"] = round(row["v1_ms"] / row["v2_ms"], 2)
        row["v3_speedup"] = round(row["v1_ms"] / row["v3_clean_ms"], 2)
        results.append(row)
        print(json.dumps(row), flush=True)

    out = Path("gpurun_out")
    out.mkdir(exist_ok=True)
    (out / "k1_ab_r02.json").write_text(json.dumps(results, indent=1))


if __name__ == "__main__":
    main()
