"""Dispatch-floor microbenchmark: what does one kernel cost in this
environment, eager vs graph, serial vs parallel streams?"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from factorvae_amd.ops import get_extension

ext = get_extension()
dev = torch.device("cuda:0")


def timeit(fn, iters=5):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    step_t = torch.zeros(1, device=dev, dtype=torch.int32)
    NK = 1000

    # eager: NK trivial kernels
    def eager():
        for _ in range(NK):
            ext.step_inc(step_t)
    t = timeit(eager, 3)
    print(f"eager trivial: {t / NK * 1e6:.2f} us/kernel")

    # graph: NK trivial kernels on one stream
    g = torch.cuda.CUDAGraph()
    eager()
    torch.cuda.synchronize()
    with torch.cuda.graph(g):
        for _ in range(NK):
            ext.step_inc(step_t)
    t = timeit(lambda: g.replay(), 5)
    print(f"graph serial trivial: {t / NK * 1e6:.2f} us/kernel")

    # graph with 4 parallel stream branches (fork/join under capture)
    streams = [torch.cuda.Stream() for _ in range(4)]
    counters = [torch.zeros(1, device=dev, dtype=torch.int32) for _ in range(4)]
    g2 = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g2):
        main_s = torch.cuda.current_stream()
        for s in streams:
            s.wait_stream(main_s)
        for si, s in enumerate(streams):
            with torch.cuda.stream(s):
                for _ in range(NK // 4):
                    ext.step_inc(counters[si])
        for s in streams:
            main_s.wait_stream(s)
    t = timeit(lambda: g2.replay(), 5)
    print(f"graph 4-stream trivial: {t / NK * 1e6:.2f} us/kernel ({NK} kernels)")

    # torch's own trivial kernel for comparison
    buf = torch.zeros(256, device=dev)
    g3 = torch.cuda.CUDAGraph()
    for _ in range(3):
        buf.add_(1.0)
    torch.cuda.synchronize()
    with torch.cuda.graph(g3):
        for _ in range(NK):
            buf.add_(1.0)
    t = timeit(lambda: g3.replay(), 5)
    print(f"graph serial torch add_: {t / NK * 1e6:.2f} us/kernel")

    # sustained big GEMM (R=6000 x 158 -> 192), back-to-back
    R, Ci, Co = 6000, 158, 192
    A = torch.randn(R, Ci, device=dev)
    W = torch.randn(Co, Ci, device=dev)
    b = torch.randn(Co, device=dev)
    out = torch.empty(R, Co, device=dev)

    def gemms():
        for _ in range(50):
            ext.gemm_nt(A, W, b, out, 1.0, False, False)
    t = timeit(gemms, 3)
    flop = 2.0 * R * Ci * Co
    print(f"gemm_nt {R}x{Ci}x{Co}: {t / 50 * 1e6:.2f} us/call = {flop / (t / 50) / 1e12:.1f} TF")

    # GRU fwd sustained
    N, T, H = 300, 20, 64
    gi = torch.randn(N, T, 3 * H, device=dev)
    Whh = torch.randn(3 * H, H, device=dev)
    bhh = torch.randn(3 * H, device=dev)
    hf = torch.empty(N, H, device=dev)
    hs = torch.empty(N, T, H, device=dev)
    hp = torch.empty(N, T, H, device=dev)
    g4 = torch.empty(N, T, 4 * H, device=dev)

    def grus():
        for _ in range(20):
            ext.gru_fwd(gi, Whh, bhh, hf, hs, hp, g4, N, T, H)
    t = timeit(grus, 3)
    print(f"gru_fwd N={N}: {t / 20 * 1e6:.2f} us/call")

    # dual-stream big GEMM overlap test
    out2 = torch.empty(R, Co, device=dev)
    s1, s2 = torch.cuda.Stream(), torch.cuda.Stream()

    def gemms2():
        cur = torch.cuda.current_stream()
        s1.wait_stream(cur)
        s2.wait_stream(cur)
        with torch.cuda.stream(s1):
            for _ in range(25):
                ext.gemm_nt(A, W, b, out, 1.0, False, False)
        with torch.cuda.stream(s2):
            for _ in range(25):
                ext.gemm_nt(A, W, b, out2, 1.0, False, False)
        cur.wait_stream(s1)
        cur.wait_stream(s2)
    t = timeit(gemms2, 3)
    print(f"gemm_nt 2-stream: {t / 50 * 1e6:.2f} us/call-equivalent")


def gemm_bench():
    """Isolated timings of the hot GEMM shapes (no contention)."""
    shapes_nt = [(6000, 158, 192), (6000, 158, 158), (210000, 158, 192)]
    for R, Ci, Co in shapes_nt:
        A = torch.randn(R, Ci, device=dev)
        W = torch.randn(Co, Ci, device=dev)
        b = torch.randn(Co, device=dev)
        out = torch.empty(R, Co, device=dev)
        t = timeit(lambda: ext.gemm_nt(A, W, b, out, 1.0, False, False), 20)
        fl = 2.0 * R * Ci * Co
        print(f"gemm_nt {R}x{Ci}x{Co}: {t*1e6:.1f} us  {fl/t/1e12:.1f} TF/s")
    for R, Ci, Co in [(6000, 192, 158), (210000, 192, 158)]:
        A = torch.randn(R, Ci, device=dev)
        B = torch.randn(Ci, Co, device=dev)
        out = torch.empty(R, Co, device=dev)
        t = timeit(lambda: ext.gemm_nn(A, B, None, out, 1.0, False, False), 20)
        fl = 2.0 * R * Ci * Co
        print(f"gemm_nn {R}x{Ci}x{Co}: {t*1e6:.1f} us  {fl/t/1e12:.1f} TF/s")
    for R, Ci, Co in [(6000, 158, 192), (210000, 158, 192)]:
        A = torch.randn(R, Ci, device=dev).bfloat16()
        W = torch.randn(Co, Ci, device=dev).bfloat16()
        b = torch.randn(Co, device=dev)
        out = torch.empty(R, Co, device=dev)
        t = timeit(lambda: ext.gemm_nt_bf16(A, W, b, out, None, 1.0, False, False), 20)
        fl = 2.0 * R * Ci * Co
        print(f"gemm_nt_bf16 {R}x{Ci}x{Co}: {t*1e6:.1f} us  {fl/t/1e12:.1f} TF/s")
    for R, Ci, Co in [(6000, 192, 158), (210000, 192, 158)]:
        A = torch.randn(R, Ci, device=dev).bfloat16()
        B = torch.randn(Ci, Co, device=dev).bfloat16()
        out = torch.empty(R, Co, device=dev, dtype=torch.bfloat16)
        t = timeit(lambda: ext.gemm_nn_bf16(A, B, None, None, out, 1.0, False, False), 20)
        fl = 2.0 * R * Ci * Co
        print(f"gemm_nn_bf16 {R}x{Ci}x{Co}: {t*1e6:.1f} us  {fl/t/1e12:.1f} TF/s")
    for R, Ci, Co in [(6000, 158, 192), (210000, 158, 192), (210000, 192, 158)]:
        KP = (Ci + 31) & ~31
        Abuf = torch.zeros(R * Ci + 8, device=dev, dtype=torch.bfloat16)
        Abuf[:R * Ci] = torch.randn(R * Ci, device=dev).bfloat16()
        A = Abuf[:R * Ci].view(R, Ci)
        Wp = torch.zeros(Co, KP, device=dev, dtype=torch.bfloat16)
        Wp[:, :Ci] = torch.randn(Co, Ci, device=dev).bfloat16()
        out = torch.empty(R, Co, device=dev, dtype=torch.bfloat16)
        t = timeit(lambda: ext.gemm_nt_bf16_rs(A, Wp, None, None, out, None, 1.0, False), 20)
        fl = 2.0 * R * Ci * Co
        print(f"gemm_nt_bf16_rs {R}x{Ci}x{Co}: {t*1e6:.1f} us  {fl/t/1e12:.1f} TF/s")
    for R, M, N, chunks in [(6000, 192, 158, 8), (210000, 192, 158, 8)]:
        A = torch.randn(R, M, device=dev).bfloat16()
        B = torch.randn(R, N, device=dev).bfloat16()
        out = torch.zeros(M, N, device=dev)
        db = torch.zeros(M, device=dev)
        part = torch.zeros(128 * M * N, device=dev)
        db_part = torch.zeros(128 * M, device=dev)
        t = timeit(lambda: ext.gemm_tn_bf16(A, B, out, part, chunks, True, db, db_part), 20)
        fl = 2.0 * R * M * N
        print(f"gemm_tn_bf16 {R}x{M}x{N} z{chunks}+bias: {t*1e6:.1f} us  {fl/t/1e12:.1f} TF/s")
    for R, M, N, chunks in [(6000, 192, 158, 8), (210000, 192, 158, 8)]:
        A = torch.randn(R, M, device=dev)
        B = torch.randn(R, N, device=dev)
        out = torch.zeros(M, N, device=dev)
        db = torch.zeros(M, device=dev)
        part = torch.zeros(32 * M * N, device=dev)
        db_part = torch.zeros(32 * M, device=dev)
        t = timeit(lambda: ext.gemm_tn(A, B, out, part, chunks, True, db, db_part), 20)
        fl = 2.0 * R * M * N
        print(f"gemm_tn {R}x{M}x{N} z{chunks}+bias: {t*1e6:.1f} us  {fl/t/1e12:.1f} TF/s")


if __name__ == "__main__":
    import os as _os
    if _os.environ.get("MICROBENCH_ONLY") == "gemm":
        gemm_bench()
    else:
        main()
        gemm_bench()
