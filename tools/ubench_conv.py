"""Per-shape conv microbenchmark: TFLOP/s for fwd / dgrad / wgrad on the
model's real layer shapes.  Run on an MI355X box."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from sodkit import ops  # noqa: E402

# (name, N, Cin, H, Cout, k, stride, pad, dil)
SHAPES = [
    ("stem7x7",    32, 3, 352, 64, 7, 2, 3, 1),
    ("l1_1x1r",    32, 256, 88, 64, 1, 1, 0, 1),
    ("l1_3x3",     32, 64, 88, 64, 3, 1, 1, 1),
    ("l1_1x1e",    32, 64, 88, 256, 1, 1, 0, 1),
    ("l2_3x3",     32, 128, 44, 128, 3, 1, 1, 1),
    ("l2_1x1r",    32, 512, 44, 128, 1, 1, 0, 1),
    ("l2_1x1e",    32, 128, 44, 512, 1, 1, 0, 1),
    ("l3_3x3",     32, 256, 22, 256, 3, 1, 1, 1),
    ("l3_1x1r",    32, 1024, 22, 256, 1, 1, 0, 1),
    ("l3_1x1e",    32, 256, 22, 1024, 1, 1, 0, 1),
    ("l4_3x3",     32, 512, 11, 512, 3, 1, 1, 1),
    ("l4_1x1r",    32, 2048, 11, 512, 1, 1, 0, 1),
    ("l4_1x1e",    32, 512, 11, 2048, 1, 1, 0, 1),
    ("dec_3x3_176", 32, 64, 176, 64, 3, 1, 1, 1),
    ("dec_3x3_88", 32, 64, 88, 64, 3, 1, 1, 1),
    ("trans_1x1",  32, 2048, 11, 64, 1, 1, 0, 1),
]


def bench(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = "cuda"
    print(f"{'shape':>12} {'M':>8} {'K':>6} {'N':>6} | {'fwd_tf':>7} {'dgrad_tf':>8} {'wgrad_tf':>8} | {'fwd8p':>7} {'dgrad8p':>7} {'wdir':>7} {'wk1':>7} {'fdir':>7} {'ddir':>7} wg8")
    for name, N, Cin, H, Cout, k, st, pad, dil in SHAPES:
        x = torch.randn(N, Cin, H, H, device=dev).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        w = (torch.randn(Cout, k, k, Cin, device=dev) * 0.1).to(torch.bfloat16)
        from sodkit import _C
        P = (H + 2 * pad - dil * (k - 1) - 1) // st + 1
        M = N * P * P
        K = k * k * Cin
        flop = 2.0 * M * K * Cout
        nob = torch.Tensor()
        t_f = bench(lambda: _C.conv2d_fwd(x, w, nob, st, pad, dil, Cin))
        y = _C.conv2d_fwd(x, w, nob, st, pad, dil, Cin)
        w_t = w.permute(3, 1, 2, 0).contiguous()
        t_d = bench(lambda: _C.conv2d_dgrad(y, w_t, H, H, st, pad, dil, Cout))
        t_w = bench(lambda: _C.conv2d_wgrad(x, y, k, k, st, pad, dil, Cin, Cout))
        # 8-phase backend A/B where the shape is eligible (wg8 = its launch
        # size; the dispatch gate needs >= ~256 to fill the chip)
        f8 = d8 = "      -"
        wg8 = ((M + 255) // 256) * (Cout // 256)
        if Cout % 256 == 0 and Cin % 8 == 0:
            t = bench(lambda: _C.conv2d_fwd_8p(x, w, nob, st, pad, dil, Cin))
            f8 = f"{flop / t / 1e12:7.0f}"
        if Cin % 256 == 0 and Cout % 8 == 0:
            t = bench(lambda: _C.conv2d_dgrad_8p(y, w_t, H, H, st, pad, dil,
                                                 Cout))
            d8 = f"{flop / t / 1e12:7.0f}"
        wd = "      -"
        if _C.wgrad_dir_eligible(k, k, st, pad, dil, Cin, H):
            t = bench(lambda: _C.conv2d_wgrad_dir(x, y, Cin, Cout))
            wd = f"{flop / t / 1e12:7.0f}"
        wk1 = "      -"
        if k == 1 and pad == 0 and dil == 1 and _C.wgrad_k1_eligible(
                1, 1, 1, 0, Cin, Cout):
            t = bench(lambda: _C.conv2d_wgrad_k1(x, y, st, Cin, Cout))
            wk1 = f"{flop / t / 1e12:7.0f}"
        fdir = ddir = "      -"
        if _C.conv_dir_eligible(k, k, st, pad, dil, Cin, Cout, H):
            t = bench(lambda: _C.conv2d_fwd_dir(x, w, nob, Cin))
            fdir = f"{flop / t / 1e12:7.0f}"
        if _C.conv_dir_eligible(k, k, st, pad, dil, Cout, Cin, H):
            t = bench(lambda: _C.conv2d_dgrad_dir(y, w_t, Cout))
            ddir = f"{flop / t / 1e12:7.0f}"
        print(f"{name:>12} {M:>8} {K:>6} {Cout:>6} | "
              f"{flop / t_f / 1e12:7.0f} {flop / t_d / 1e12:8.0f} "
              f"{flop / t_w / 1e12:8.0f} | {f8} {d8} {wd} {wk1} {fdir} {ddir} wg8={wg8}")


if __name__ == "__main__":
    main()
