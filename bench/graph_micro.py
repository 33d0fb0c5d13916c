"""hipGraph replay-cost micro-benchmark: CPU enqueue cost and end-to-end
cost vs node count (informs whether merging graph nodes in the scoring
session pays — the serving step's largest host cost is the ~50-65 us
graph launch)."""

import time

import torch


def main():
    assert torch.cuda.is_available()
    torch.cuda.init()
    s = torch.cuda.Stream()
    print("nodes  enqueue_us  e2e_us")
    for n_nodes in (1, 2, 4, 8, 12, 16, 24):
        x = torch.zeros(256, device="cuda")
        with torch.cuda.stream(s):
            for _ in range(3):  # warm kernels
                x.add_(1)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, stream=s):
            for _ in range(n_nodes):
                x.add_(1)
        for _ in range(20):
            g.replay()
        torch.cuda.synchronize()
        N = 2000
        t0 = time.perf_counter()
        for _ in range(N):
            g.replay()
        t_enq = (time.perf_counter() - t0) / N * 1e6  # pure CPU enqueue if GPU keeps up
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(N):
            g.replay()
        torch.cuda.synchronize()
        t_e2e = (time.perf_counter() - t0) / N * 1e6
        print(f"{n_nodes:5d}  {t_enq:9.2f}  {t_e2e:7.2f}")


if __name__ == "__main__":
    main()
