"""DistillReader throughput tool (parity: reference
example/distill/qps_tools/distill_reader_qps.py — steps/s with a synthetic
reader and the NOP teacher)."""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import edl_amd.distill.worker as W  # noqa: E402
from edl_amd.distill.reader import DistillReader  # noqa: E402


def main(n_batches=2000, batch=32, require_num=4):
    W._NOP_PREDICT_TEST = True

    def gen():
        x = np.zeros((batch, 8), dtype=np.float32)
        y = np.zeros((batch,), dtype=np.int64)
        for _ in range(n_batches):
            yield (x, y)

    dr = DistillReader(["x", "y"], ["p"], require_num=require_num)
    dr.set_batch_generator(gen)
    dr.set_fixed_teacher(["nop:0"])
    t0 = time.monotonic()
    n = sum(1 for _ in dr())
    dt = time.monotonic() - t0
    print("%d batches in %.2fs = %.0f batches/s (%.0f samples/s)" % (
        n, dt, n / dt, n * batch / dt))


if __name__ == "__main__":
    main()
