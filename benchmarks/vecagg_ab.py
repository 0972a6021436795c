"""MFMA vs VALU for vector aggregates (colsum of [N,16] f32).

The op is 0.25 FLOP/byte: HBM-bound, so matrix cores cannot beat a
plain accumulation loop — this measurement is the evidence behind
using the rocPRIM reduce-by-key path (not MFMA) for production vector
aggregates.  Run on an MI355X box.
"""
import sys, time
import torch
sys.path.insert(0, ".")
from bigslice_amd.kernels import _C

N = 64_000_000
x = torch.randn(N, 16, dtype=torch.float32, device="cuda")
ref = x.to(torch.float64).sum(0)
for mfma in (False, True):
    out = _C.colsum16(x, mfma)
    err = (out.to(torch.float64) - ref).abs().max().item()
    rel = err / ref.abs().max().item()
    for _ in range(2): _C.colsum16(x, mfma)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): _C.colsum16(x, mfma)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 10 * 1000
    gbs = N * 16 * 4 / ms / 1e6
    print(f"{'mfma' if mfma else 'valu'}: {ms:6.3f} ms  {gbs:7.1f} GB/s"
          f"  max rel err {rel:.2e}")
