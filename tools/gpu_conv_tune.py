"""On-GPU sweep of conv-kernel variants (WAVES, RPW) vs MIOpen.

Writes timings to stdout; pick winners into kLaunches (conv_mfma.hip)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

DEV = "cuda:0"
SHAPES = [
    ("conv1-u8", 84, 84, 4, 8, 8, 4, 32, True, [0, 1, 2, 3]),
    ("conv2", 20, 20, 32, 4, 4, 2, 64, False, [4, 5, 6, 7]),
    ("conv3", 9, 9, 64, 3, 3, 1, 64, False, [8, 9, 10, 11]),
]
VNAMES = {0: "w4r16", 1: "w8r16", 2: "w4r32", 3: "w8r32"}


def timeit(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    from distributed_rl_amd.ops import hip_ext

    ext = hip_ext(required=True)
    N = 512
    for name, H, W, C, KH, KW, S, COUT, u8, variants in SHAPES:
        P, Q = (H - KH) // S + 1, (W - KW) // S + 1
        flops = 2.0 * N * P * Q * COUT * KH * KW * C
        if u8:
            x = torch.randint(0, 256, (N, C, H, W), dtype=torch.uint8,
                              device=DEV).to(memory_format=torch.channels_last)
        else:
            x = torch.randn(N, C, H, W, device=DEV).to(torch.bfloat16).to(
                memory_format=torch.channels_last)
        w = (torch.randn(COUT, C, KH, KW, device=DEV) * 0.05).to(
            torch.bfloat16).to(memory_format=torch.channels_last)
        b = torch.zeros(COUT, device=DEV, dtype=torch.bfloat16)
        out = torch.empty(N, COUT, P, Q, dtype=torch.bfloat16, device=DEV).to(
            memory_format=torch.channels_last)
        print(f"== {name}: M={N*P*Q} K={KH*KW*C} N={COUT} ({flops/1e9:.2f} GF)")
        for v in variants:
            ok = ext.conv_fwd_variant(x, w, b, out, S, v)
            assert ok, (name, v)
            us = timeit(lambda: ext.conv_fwd_variant(x, w, b, out, S, v))
            print(f"  {VNAMES[v % 4]:6s}: {us:7.1f} us  {flops/us/1e6:7.1f} TF")
        if not u8:
            xf = x
            ref_us = timeit(lambda: F.conv2d(xf, w, b, stride=S))
            print(f"  miopen: {ref_us:7.1f} us  {flops/ref_us/1e6:7.1f} TF")




def bench_wrw():
    import torch.nn.functional as F
    from distributed_rl_amd.ops import hip_ext

    ext = hip_ext(required=True)
    N = 512
    for name, H, W, C, KH, KW, S, COUT, u8, _ in SHAPES:
        P, Q = (H - KH) // S + 1, (W - KW) // S + 1
        flops = 2.0 * N * P * Q * COUT * KH * KW * C
        if u8:
            x = torch.randint(0, 256, (N, C, H, W), dtype=torch.uint8,
                              device=DEV).to(memory_format=torch.channels_last)
        else:
            x = torch.randn(N, C, H, W, device=DEV).to(torch.bfloat16).to(
                memory_format=torch.channels_last)
        gout = torch.randn(N, COUT, P, Q, device=DEV).to(torch.bfloat16).to(
            memory_format=torch.channels_last)
        ws = torch.zeros(COUT, KH * KW * C, dtype=torch.float32, device=DEV)
        us = timeit(lambda: (ws.zero_(), ext.conv_wrw(x, gout, ws, torch.empty(0, device=DEV), S)))
        line = f"wrw {name}: {us:7.1f} us  {flops/us/1e6:7.1f} TF"
        if not u8:
            w = torch.empty(COUT, C, KH, KW, device=DEV, dtype=torch.bfloat16
                            ).to(memory_format=torch.channels_last)
            ref = timeit(lambda: torch.ops.aten.convolution_backward(
                gout, x, w, None, [S, S], [0, 0], [1, 1], False, [0, 0], 1,
                [False, True, False]))
            line += f"   (miopen {ref:7.1f} us {flops/ref/1e6:6.1f} TF)"
        print(line)


if __name__ == "__main__":
    main()
    bench_wrw()
