"""Microbench: own conv_dgrad vs aten (MIOpen) per geometry, plus the
per-step vs persistent LSTM kernels — decides hot-path routing."""

import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributed_rl_amd.ops import hip_ext

DEV = "cuda:0"


def t_ms(fn, iters=50, warm=10):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    ext = hip_ext(required=True)
    for (H, W, C, KH, KW, S, COUT, N) in [
        (20, 20, 32, 4, 4, 2, 64, 512),
        (9, 9, 64, 3, 3, 1, 64, 512),
        (20, 20, 32, 4, 4, 2, 64, 2560),
        (9, 9, 64, 3, 3, 1, 64, 2560),
    ]:
        P, Q = (H - KH) // S + 1, (W - KW) // S + 1
        gout = (torch.randn(N, COUT, P, Q, device=DEV) * .5).to(
            torch.bfloat16).contiguous(memory_format=torch.channels_last)
        w = (torch.randn(COUT, C, KH, KW, device=DEV) * .1).to(
            torch.bfloat16).contiguous(memory_format=torch.channels_last)
        dx = torch.empty(N, C, H, W, dtype=torch.bfloat16, device=DEV
                         ).contiguous(memory_format=torch.channels_last)
        w_t = torch.empty(w.numel(), dtype=torch.bfloat16, device=DEV)
        x_d = torch.zeros_like(dx)
        own = t_ms(lambda: ext.conv_dgrad(gout, w, w_t, dx, S))
        aten = t_ms(lambda: torch.ops.aten.convolution_backward(
            gout, x_d, w, None, [S, S], [0, 0], [1, 1], False, [0, 0], 1,
            [True, False, False]))
        print(f"dgrad {H}x{W}x{C} k{KH} s{S} N={N}: own {own*1e3:.1f}us "
              f"aten {aten*1e3:.1f}us")

    # LSTM per-step vs persistent (fwd+bwd through manual_lstm_seq)
    from distributed_rl_amd.models.manual_lstm import manual_lstm_seq

    lstm = torch.nn.LSTM(3136, 512).to(DEV)
    x = torch.randn(80, 32, 3136, device=DEV).to(torch.bfloat16)
    h0 = torch.randn(1, 32, 512, device=DEV)
    c0 = torch.randn(1, 32, 512, device=DEV)

    def run():
        xg = x.detach().requires_grad_(True)
        out, _ = manual_lstm_seq(xg, (h0, c0), lstm)
        out.backward(torch.ones_like(out))

    for flag in ("1", "0"):
        os.environ["DRL_LSTM_PERSISTENT_BF16"] = flag
        ms = t_ms(run, iters=20, warm=5)
        print(f"lstm seq80 fwd+bwd persistent={flag}: {ms:.3f} ms")


if __name__ == "__main__" and os.environ.get("DRL_DIAG") != "1":
    main()


def lstm_diag():
    ext = hip_ext(required=True)
    B, H, T = 32, 512, 80
    h = torch.randn(2, B, H, device=DEV).to(torch.bfloat16)
    w = torch.randn(4 * H, H, device=DEV).to(torch.bfloat16)
    sink = torch.zeros(32, device=DEV)
    for mode, name in [(0, "barrier-only"), (1, "+h-stage"), (2, "+mfma")]:
        def run():
            ctr = torch.zeros(1, dtype=torch.int32, device=DEV)
            ext.lstm_diag(h, w, sink, ctr, B, T, mode)
        ms = t_ms(run, iters=30, warm=5)
        print(f"lstm_diag {name}: {ms*1e3:7.1f} us total, "
              f"{ms*1e3/T:6.2f} us/step")


if __name__ == "__main__" and os.environ.get("DRL_DIAG") == "1":
    lstm_diag()
