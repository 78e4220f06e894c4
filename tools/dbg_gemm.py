import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, cake_amd
from tests.helpers import quantize_bf16
for (M, N, K) in [(256, 384, 96), (200, 300, 40), (256, 384, 64)]:
    rng = np.random.default_rng(M * 7 + N)
    x = rng.standard_normal((M, K)).astype(np.float32) * 0.5
    w = rng.standard_normal((N, K)).astype(np.float32) * 0.5
    got = cake_amd.op_linear(x, w)
    ref = quantize_bf16(x) @ quantize_bf16(w).T
    err = np.abs(got - ref)
    rel = err.max() / max(1e-9, np.abs(ref).max())
    bi = np.unravel_index(err.argmax(), err.shape)
    bad_rows = np.unique(np.where(err > 0.1 * err.max())[0])
    bad_cols = np.unique(np.where(err > 0.1 * err.max())[1])
    print(f"({M},{N},{K}): rel={rel:.2e} worst at {bi} "
          f"got={got[bi]:.4f} ref={ref[bi]:.4f} "
          f"badrows[{bad_rows.min() if len(bad_rows) else -1}..{bad_rows.max() if len(bad_rows) else -1}]({len(bad_rows)}) "
          f"badcols[{bad_cols.min() if len(bad_cols) else -1}..{bad_cols.max() if len(bad_cols) else -1}]({len(bad_cols)})", flush=True)
