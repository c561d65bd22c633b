"""One-off deep differential sweep on a GPU box: many random join/agg
configs, HIP vs oracle (reuses the committed fuzz helpers). Not a test —
a bug-shaker for spare GPU minutes; exits nonzero on first mismatch."""
import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import numpy as np  # noqa: E402
import test_gpu_fuzz as F  # noqa: E402

N = int(sys.argv[1]) if len(sys.argv) > 1 else 200
BASE = int(sys.argv[2]) if len(sys.argv) > 2 else 100000
fails = 0
for seed in range(BASE, BASE + N):
    for fn in (F.test_fuzz_join, F.test_fuzz_agg, F.test_fuzz_join_conditions,
               F.test_fuzz_f64_minmax_nan_signed_zero,
               F.test_fuzz_f64_join_conditions_nan,
               F.test_fuzz_groupjoin_window):
        try:
            fn.__wrapped__(seed) if hasattr(fn, "__wrapped__") else fn(seed)
        except AssertionError:
            print(f"MISMATCH {fn.__name__} seed {seed}")
            traceback.print_exc()
            fails += 1
            if fails > 3:
                sys.exit(1)
        except Exception:
            print(f"ERROR {fn.__name__} seed {seed}")
            traceback.print_exc()
            sys.exit(2)
    if (seed - BASE) % 25 == 0:
        print(f"... {seed - BASE}/{N}", flush=True)
print("DEEP FUZZ:", "FAILED" if fails else "CLEAN", N, "seeds x 6 suites")
sys.exit(1 if fails else 0)
