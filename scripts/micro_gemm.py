#!/usr/bin/env python3
"""Per-shape GEMM timings for the wide-and-deep MLP (fwd/dgrad/wgrad),
bf16, batch 65536 — identifies which hipBLASLt shapes underperform."""

import os
import sys
import time

os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")

import torch

B = 65536
LAYERS = [(432, 1024), (1024, 512), (512, 256)]


def timeit(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    for (cin, cout) in LAYERS:
        x = torch.randn(B, cin, device="cuda").to(torch.bfloat16)
        w = torch.randn(cout, cin, device="cuda").to(torch.bfloat16)
        dy = torch.randn(B, cout, device="cuda").to(torch.bfloat16)
        t_fwd = timeit(lambda: x.matmul(w.t()))
        t_dgrad = timeit(lambda: dy.matmul(w))
        t_wgrad = timeit(lambda: dy.t().matmul(x))
        gf = 2 * B * cin * cout / 1e9
        print(f"[{cin:>5}->{cout:>4}] "
              f"fwd {t_fwd:7.1f}us ({gf/t_fwd*1e3:5.0f} TF)  "
              f"dgrad {t_dgrad:7.1f}us ({gf/t_dgrad*1e3:5.0f} TF)  "
              f"wgrad {t_wgrad:7.1f}us ({gf/t_wgrad*1e3:5.0f} TF)")
    # head GEMV shapes
    x = torch.randn(B, 256, device="cuda").to(torch.bfloat16)
    w1 = torch.randn(256, device="cuda").to(torch.bfloat16)
    dy = torch.randn(B, device="cuda").to(torch.bfloat16)
    t_mv = timeit(lambda: x @ w1)
    t_wv = timeit(lambda: x.t() @ dy)
    print(f"[head 256] x@w {t_mv:7.1f}us   x.t()@dy {t_wv:7.1f}us")
    # wgrad in fp32 output (hipblaslt C dtype fp32)
    for (cin, cout) in LAYERS:
        x = torch.randn(B, cin, device="cuda").to(torch.bfloat16)
        dy = torch.randn(B, cout, device="cuda").to(torch.bfloat16)
        t = timeit(lambda: torch.mm(dy.t().float(), x.float())) \
            if False else None
        # bf16 input fp32 out via addmm out_dtype not exposed; instead try
        # chunked K: two half-batch wgrads summed
        half = B // 2
        t_chunk = timeit(lambda: dy[:half].t().matmul(x[:half])
                         + dy[half:].t().matmul(x[half:]))
        gf = 2 * B * cin * cout / 1e9
        print(f"[wgrad chunk2 {cin}x{cout}] {t_chunk:7.1f}us "
              f"({gf/t_chunk*1e3:5.0f} TF)")


def wgrad_b_sweep():
    """Is wgrad128 bound by redundant HBM traffic or by its inner
    structure?  At B=8192 both operands fit in L2/L3 -> if the TF rate
    jumps, traffic is the bound; if flat, the loop structure is."""
    import tf_yarn_amd.ops._C as C
    for Bs in (8192, 16384, 65536):
        dy = torch.randn(Bs, 1024, device="cuda").to(torch.bfloat16)
        x = torch.randn(Bs, 432, device="cuda").to(torch.bfloat16)
        gf = 2 * Bs * 1024 * 432 / 1e9
        t_lib = timeit(lambda: dy.t().matmul(x))
        best = None
        for sk in (4, 8, 16, 32):
            t = timeit(lambda: C.wgrad_nt128(dy, x, sk))
            best = t if best is None or t < best else best
        print(f"[B={Bs:6}] lib {t_lib:7.1f}us ({gf/t_lib*1e3:5.0f} TF)  "
              f"wgrad128 {best:7.1f}us ({gf/best*1e3:5.0f} TF)")


def custom_wgrad():
    import tf_yarn_amd.ops._C as C
    for (cin, cout) in LAYERS_PAD:
        x = torch.randn(B, cin, device="cuda").to(torch.bfloat16)
        dy = torch.randn(B, cout, device="cuda").to(torch.bfloat16)
        for sk in (2, 4, 8, 16):
            t = timeit(lambda: C.wgrad_nt(dy, x, sk))
            gf = 2 * B * cin * cout / 1e9
            print(f"[custom wgrad {cout}x{cin} sk={sk:2}] {t:7.1f}us "
                  f"({gf/t*1e3:5.0f} TF)")
    for (cin, cout) in [(432, 1024)] + LAYERS_PAD[1:]:
        x = torch.randn(B, cin, device="cuda").to(torch.bfloat16)
        dy = torch.randn(B, cout, device="cuda").to(torch.bfloat16)
        for sk in (4, 8, 16, 32):
            t = timeit(lambda: C.wgrad_nt128(dy, x, sk))
            gf = 2 * B * cin * cout / 1e9
            print(f"[wgrad128 {cout}x{cin} sk={sk:2}] {t:7.1f}us "
                  f"({gf/t*1e3:5.0f} TF)")
    for (cin, cout) in [(432, 1024), (1024, 512), (512, 256)]:
        x = torch.randn(B, cin, device="cuda").to(torch.bfloat16)
        dy = torch.randn(B, cout, device="cuda").to(torch.bfloat16)
        gf = 2 * B * cin * cout / 1e9
        for sk in (24, 28, 32, 36, 40, 48):
            try:
                t = timeit(lambda: C.wgrad_nt256(dy, x, sk))
                print(f"[wgrad256 {cout}x{cin} sk={sk:2}] {t:7.1f}us "
                      f"({gf/t*1e3:5.0f} TF)")
            except RuntimeError as e:
                print(f"[wgrad256 {cout}x{cin}] skipped: {e}")
                break
    x = torch.randn(B, 256, device="cuda").to(torch.bfloat16)
    dy = torch.randn(B, device="cuda").to(torch.bfloat16)
    t = timeit(lambda: C.col_reduce_dot(x, dy))
    print(f"[head col_reduce_dot 256] {t:7.1f}us")


def custom_fwd():
    """gemm_bt (custom fused fwd GEMM + bias/ReLU epilogue) vs the lib
    pipeline (hipBLASLt x@w.t() then the bias_relu kernel) on the MLP
    forward shapes, and vs plain dy@w on the dgrad shapes."""
    import tf_yarn_amd.ops._C as C
    print("== forward: z=x@w.t(); y=relu(z+b)  vs  gemm_bt(..., relu) ==")
    for (cin, cout) in LAYERS:
        x = torch.randn(B, cin, device="cuda").to(torch.bfloat16)
        w = torch.randn(cout, cin, device="cuda").to(torch.bfloat16) * 0.03
        bias = torch.randn(cout, device="cuda").to(torch.bfloat16)
        # numerics: vs fp32 reference
        ref = torch.relu(x.float() @ w.float().t() + bias.float())
        out = C.gemm_bt(x, w, bias, True)
        err = (out.float() - ref).abs().max().item()
        rel = err / ref.abs().max().clamp_min(1e-6).item()
        gf = 2 * B * cin * cout / 1e9
        t_lib = timeit(lambda: C.bias_relu_fwd(x.matmul(w.t()), bias))
        t_cus = timeit(lambda: C.gemm_bt(x, w, bias, True))
        print(f"[fwd {cin:>5}->{cout:>4}] lib {t_lib:7.1f}us "
              f"({gf/t_lib*1e3:5.0f} TF)  gemm_bt {t_cus:7.1f}us "
              f"({gf/t_cus*1e3:5.0f} TF)  relerr {rel:.3e}")
    print("== dgrad: dy@w  vs  gemm_bt(dy, w.t().contig) ==")
    for (cin, cout) in LAYERS:
        dy = torch.randn(B, cout, device="cuda").to(torch.bfloat16)
        w = torch.randn(cout, cin, device="cuda").to(torch.bfloat16) * 0.03
        wt = w.t().contiguous()
        ref = dy.float() @ w.float()
        out = C.gemm_bt(dy, wt, None, False)
        rel = ((out.float() - ref).abs().max()
               / ref.abs().max().clamp_min(1e-6)).item()
        gf = 2 * B * cin * cout / 1e9
        t_lib = timeit(lambda: dy.matmul(w))
        t_cus = timeit(lambda: C.gemm_bt(dy, wt, None, False))
        t_cus_t = timeit(lambda: C.gemm_bt(dy, w.t().contiguous(),
                                           None, False))
        print(f"[dgrad {cout:>4}->{cin:>5}] lib {t_lib:7.1f}us "
              f"({gf/t_lib*1e3:5.0f} TF)  gemm_bt {t_cus:7.1f}us "
              f"({gf/t_cus*1e3:5.0f} TF)  +transpose {t_cus_t:7.1f}us  "
              f"relerr {rel:.3e}")


LAYERS_PAD = [(448, 1024), (1024, 512), (512, 256)]

if __name__ == "__main__":
    assert torch.cuda.is_available()
    import sys as _sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    if "--bsweep" in _sys.argv:
        wgrad_b_sweep()
    elif "--fwd" in _sys.argv:
        custom_fwd()
    elif "--custom" in _sys.argv:
        custom_wgrad()
    else:
        main()
        custom_wgrad()
