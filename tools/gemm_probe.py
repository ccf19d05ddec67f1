import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import zaremba_amd._hip as ext

dev = "cuda"
def bench(M, N, K, out_dtype=torch.float32, iters=30):
    A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    B = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    C = torch.empty(M, N, device=dev, dtype=out_dtype)
    for _ in range(5): ext.gemm(A, B, C, None, False, False)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): ext.gemm(A, B, C, None, False, False)
    torch.cuda.synchronize(); dt = (time.perf_counter()-t0)/iters
    tf = 2*M*N*K/dt/1e12
    print(f"NT M={M:6d} N={N:6d} K={K:6d} {str(out_dtype)[6:]:9s}: "
          f"{dt*1e6:8.1f} us  {tf:7.1f} TF")

# dW shapes: measured in-train ~56us
bench(6000, 1500, 700)
bench(6000, 1500, 704)   # no K-tail, 11 tiles
bench(6016, 1536, 768)   # all-interior variant
bench(6000, 1500, 768)   # K padded only
bench(10000, 1500, 700)
bench(10048, 1536, 768)
# dx shapes (bf16 out): in-train ~88us
bench(700, 1500, 6000, torch.bfloat16)
bench(768, 1536, 6016, torch.bfloat16)
bench(700, 1500, 10000, torch.bfloat16)
# input gemm shapes
bench(700, 6000, 1500, torch.bfloat16)
bench(768, 6016, 1536, torch.bfloat16)
# proj
bench(700, 10000, 1500)
bench(768, 10048, 1536)

# --- split-K hypothesis probe: dx GEMMs run ~1 block/CU (264 WGs of
# 64-tile); if doubling the grid doesn't double time, a 2-way K-split
# (2 co-resident blocks/CU interleaving their latency chains) pays.
if len(sys.argv) > 1 and sys.argv[1] == "splitk":
    print("== grid-doubling probe (64-tile dx shapes) ==")
    bench(700, 1500, 6016, torch.bfloat16)    # 264 WGs
    bench(1400, 1500, 6016, torch.bfloat16)   # 528 WGs, 2x work
    bench(2800, 1500, 6016, torch.bfloat16)   # 1056 WGs, 4x work
    bench(700, 3000, 6016, torch.bfloat16)    # 528 WGs via N
    bench(700, 1500, 3008, torch.bfloat16)    # half-K: lower bound/2way
    bench(700, 1500, 10048, torch.bfloat16)   # proj dx
    bench(700, 1500, 5024, torch.bfloat16)    # its half-K
    bench(1400, 1500, 10048, torch.bfloat16)

if len(sys.argv) > 1 and sys.argv[1] == "splitk2":
    def bench_splitk(M, N, K, k_pad=0, iters=30):
        Keff = k_pad if k_pad else K
        A = torch.randn(M, Keff, device=dev, dtype=torch.bfloat16)[:, :K].contiguous()
        flat = torch.zeros(M*K + 64, device=dev, dtype=torch.bfloat16)
        flat[:M*K].copy_(A.reshape(-1)); A = flat[:M*K].view(M, K)
        B = torch.zeros(N, Keff, device=dev, dtype=torch.bfloat16); B[:, :K].normal_()
        C1 = torch.empty(M, N, device=dev, dtype=torch.float32)
        Cex = torch.empty(3, M, N, device=dev, dtype=torch.float32)
        out = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        for _ in range(5):
            nz = ext.gemm_splitk(A, B, C1, Cex, None, k_pad); ext.addn_f32_bf16(C1, Cex, out, nz)
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(iters):
            nz = ext.gemm_splitk(A, B, C1, Cex, None, k_pad); ext.addn_f32_bf16(C1, Cex, out, nz)
        torch.cuda.synchronize(); dt = (time.perf_counter()-t0)/iters
        print(f"SK M={M:6d} N={N:6d} K={K:6d} (pad {Keff}): {dt*1e6:8.1f} us  {2*M*N*K/dt/1e12:7.1f} TF")
    print("== split-K + combine vs plain ==")
    bench_splitk(700, 1500, 6000, 6016)       # lstm dx (in-train now)
    bench_splitk(700, 1500, 10000, 10048)     # proj dx
    bench_splitk(700, 6000, 1500, 1536)       # input gemm candidate
    bench(700, 6000, 1500, torch.bfloat16)    # plain comparison
    bench(1400, 6000, 1536, torch.bfloat16)   # grid-doubling check
