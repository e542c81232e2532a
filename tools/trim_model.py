#!/usr/bin/env python3
"""Strip optimizer slots from a .npy checkpoint for deployment
(parity with reference data/models/trim_model.py).

    python tools/trim_model.py <checkpoint.npy> [out.npy]
"""

import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

from sat_amd.utils.checkpoint import trim  # noqa: E402

if __name__ == '__main__':
    if len(sys.argv) < 2:
        print(__doc__)
        sys.exit(1)
    src = sys.argv[1]
    dst = sys.argv[2] if len(sys.argv) > 2 else src
    removed = trim(src, dst)
    print('removed %d optimizer entries -> %s' % (removed, dst))
