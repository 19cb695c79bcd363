"""A/B microbench of the fconv variants on the raft-things @ 55x128 shapes.

Within-process interleaved timing (guide §5.4 rules 9/24): N rounds per
variant, report median us/call. Run via gpurun.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd.ops import require_hip


def bench(fn, iters=60, rounds=7):
    times = []
    for _ in range(rounds):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        times.append((time.perf_counter() - t0) / iters * 1e6)
    times.sort()
    return times[len(times) // 2]


def main():
    hip = require_hip()
    dev = torch.device("cuda:0")
    B, H, W = 1, 55, 128
    torch.manual_seed(0)

    # (name, Cin, N, kh, kw) — the per-iteration conv set
    configs = [
        ("convc1 1x1", 328, 256, 1, 1),
        ("convc2 3x3", 256, 192, 3, 3),
        ("convf2 3x3", 128, 64, 3, 3),
        ("cv 3x3", 256, 126, 3, 3),
        ("zr1 1x5", 384, 256, 1, 5),
        ("q1 1x5", 384, 128, 1, 5),
        ("zr2 5x1", 384, 256, 5, 1),
        ("q2 5x1", 384, 128, 5, 1),
        ("heads 3x3", 128, 512, 3, 3),
        ("fh2 3x3", 256, 2, 3, 3),
        ("m2 1x1", 256, 576, 1, 1),
    ]
    total = {0: 0.0, 1: 0.0}
    print(f"{'conv':<14} {'GF':>6} "
          f"{'a0m1':>7} {'a0m2':>7} {'a1m1':>7} {'a1m2':>7} "
          f"{'big':>7} {'TF@bst':>7}")
    for name, cin, n, kh, kw in configs:
        x = torch.randn(B, H, W, cin, device=dev).to(torch.bfloat16)
        wp = torch.randn(kh * kw, n, cin, device=dev).to(torch.bfloat16) * 0.1
        bias = torch.zeros(n, device=dev)
        gf = 2.0 * B * H * W * n * cin * kh * kw / 1e9
        res = {}
        for at in (0, 1):
            for mt in (1, 2):
                res[(at, mt)] = bench(lambda: hip.fconv_plain(
                    x, None, wp, bias, kh, kw, 1, None, 0, 0, 0, at, mt,
                    1, None))
            total[at] += res[(at, 1)]
        res[("big", 1)] = bench(lambda: hip.fconv_plain(
            x, None, wp, bias, kh, kw, 1, None, 0, 0, 0, 0, -2, 1, None))
        best = min(res.values())
        cells = " ".join(f"{res[k]:>7.1f}" for k in
                         ((0, 1), (0, 2), (1, 1), (1, 2), ("big", 1)))
        print(f"{name:<14} {gf:>6.2f} {cells} {gf/best*1e6/1e3:>7.0f}")
    print(f"{'TOTAL':<14} {'':>6} {total[0]:>9.1f} {total[1]:>9.1f}")

    # GRU fused pair timing (zr + q as used by the model)
    hd, xd = 128, 256
    h = torch.randn(B, H, W, hd, device=dev).to(torch.bfloat16)
    xb = torch.randn(B, H, W, xd, device=dev).to(torch.bfloat16)
    for kh, kw, tag in ((1, 5, "horiz"), (5, 1, "vert")):
        wzr = torch.randn(kh * kw, 2 * hd, hd + xd, device=dev) \
            .to(torch.bfloat16) * 0.05
        bzr = torch.zeros(2 * hd, device=dev)
        wq = torch.randn(kh * kw, hd, hd + xd, device=dev) \
            .to(torch.bfloat16) * 0.05
        bq = torch.zeros(hd, device=dev)

        def gru_pass():
            z, rh = hip.fconv_gru_zr(h, xb, wzr, bzr, kh, kw)
            return hip.fconv_gru_q(rh, xb, wq, bq, kh, kw, z, h)

        print(f"gru {tag}: {bench(gru_pass):.1f} us/pass")

    # small-Cin specialized kernels (dispatch selected by env at import —
    # rerun the script under RAFT_AMD_SMALLC_MFMA=0 / RAFT_AMD_STEM=0 to
    # time the fallback paths for the same shapes)
    print("-- small-Cin kernels (env-dispatched; see RAFT_AMD_* switches)")
    for name, cin, n, kh, kw, slice_dim in (
            ("convf1 7x7/C2", 2, 128, 7, 7, 256),
            ("small f1 7x7/C2", 2, 64, 7, 7, 128)):
        xs = torch.randn(B, H, W, slice_dim, device=dev).to(torch.bfloat16)
        wp = torch.randn(kh * kw, n, cin, device=dev).to(torch.bfloat16) * .1
        bias = torch.zeros(n, device=dev)
        t = bench(lambda: hip.fconv_smallk(xs, wp, bias, kh, kw, 1,
                                           slice_dim - cin, cin, 1))
        print(f"{name:<18} {t:>7.1f} us")
    # stem: 7x7 stride-2 over the 8-channel padded image (fnet batch 2)
    Hs, Ws = 220, 512
    x8 = torch.randn(2, 2 * Hs, 2 * Ws, 8, device=dev).to(torch.bfloat16)
    wp8 = torch.randn(49, 64, 8, device=dev).to(torch.bfloat16) * 0.1
    b8 = torch.zeros(64, device=dev)
    t = bench(lambda: hip.fconv_plain(x8, None, wp8, b8, 7, 7, 1, None,
                                      0, 0, 0, -1, -1, 2, None), iters=20)
    print(f"{'stem 7x7 S2/C8':<18} {t:>7.1f} us")


if __name__ == "__main__":
    main()
