"""MFMA-assisted vs scalar ziggurat moment kernels."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca

N = 1 << 30
for dist in ("std_normal", "std_exponential"):
    for tag, mfma in (("scalar", 0), ("mfma  ", 1)):
        r = ca._C.rng_moments_gpu(dist=dist, n=N, seed=7, device=0, mfma=mfma)
        gsps = N / (r["elapsed_ms"] * 1e-3) / 1e9
        print(f"{dist:16s} {tag}: {gsps:7.1f} G samples/s  mean={r['mean']:+.6f} var={r['var']:.6f}", flush=True)
