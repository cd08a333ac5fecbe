"""GPU microbench: decode GEMV variants vs rocBLAS on the 8B shapes.
Within-process interleaved A/B (guide rule 24)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.nn.functional as F

ITERS = int(os.environ.get("GEMV_ITERS", "200"))


def timeit(fn, iters=None, warmup=None):
    iters = iters or ITERS
    warmup = warmup if warmup is not None else max(2, ITERS // 10)
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6   # us


def main():
    from fei_amd import ops
    lib = ops.require_lib()
    dev = "cuda:0"
    shapes = [("qkv", 6144, 4096), ("o", 4096, 4096), ("down", 4096, 14336),
              ("lm_head", 128256, 4096)]
    M = 1
    print(f"{'shape':8s} {'MB':>7s} {'rocBLAS':>9s} {'gemv':>9s} {'gemv_nt':>9s}  (us; TB/s in parens)")
    for name, N, K in shapes:
        g = torch.Generator(device=dev).manual_seed(1)
        x = (torch.randn(M, K, generator=g, device=dev) * 0.1).to(torch.bfloat16)
        w = (torch.randn(N, K, generator=g, device=dev) * 0.02).to(torch.bfloat16)
        out = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        mb = N * K * 2 / 1e6

        t_roc = timeit(lambda: F.linear(x, w))
        def kv(nt):
            lib.fei_gemv(out.data_ptr(), x.data_ptr(), w.data_ptr(), M, N, K,
                         nt, torch.cuda.current_stream().cuda_stream)
        t_g = timeit(lambda: kv(0))
        t_gnt = timeit(lambda: kv(1))
        def tb(us):
            return N * K * 2 / (us * 1e-6) / 1e12
        print(f"{name:8s} {mb:7.1f} {t_roc:7.1f}({tb(t_roc):4.1f}) "
              f"{t_g:7.1f}({tb(t_g):4.1f}) {t_gnt:7.1f}({tb(t_gnt):4.1f})")
        # correctness
        ref = F.linear(x.float(), w.float())
        kv(1)
        err = (out.float() - ref).abs().max().item()
        assert err < 0.05, (name, err)
    # fused swiglu shape
    I, K = 14336, 4096
    g = torch.Generator(device=dev).manual_seed(2)
    x = (torch.randn(1, K, generator=g, device=dev) * 0.1).to(torch.bfloat16)
    wgu = (torch.randn(2 * I, K, generator=g, device=dev) * 0.02).to(torch.bfloat16)
    o2 = torch.empty(1, I, device=dev, dtype=torch.bfloat16)
    def sw():
        lib.fei_gemv_swiglu(o2.data_ptr(), x.data_ptr(), wgu.data_ptr(), 1, I, K,
                            torch.cuda.current_stream().cuda_stream)
    t = timeit(sw)
    print(f"swiglu   {2*I*K*2/1e6:7.1f} {'':9s} {t:7.1f}({2*I*K*2/(t*1e-6)/1e12:4.1f})")


if __name__ == "__main__":
    main()
