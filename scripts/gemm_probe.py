"""Quick hipBLASLt GEMM latency probe (clock/power-cap verification)."""
import sys, time
import torch

a = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
b = torch.randn_like(a)
def t():
    for _ in range(5):
        a @ b
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(30):
        a @ b
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / 30
tag = sys.argv[1] if len(sys.argv) > 1 else ""
print(f"gemm ms {tag}: {t()*1e3:.4f}")
