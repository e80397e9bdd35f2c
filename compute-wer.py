#!/usr/bin/env python3
"""Reference-compatible scorer CLI:

    python compute-wer.py REF_CAPTIONS HYP_CAPTIONS

Prints WER, ExpRate, <=1 and <=2 error rates from two caption files
("uid<TAB>tok tok ..." lines, keyed by uid)."""

import sys

from wap_amd.eval import score_files

if __name__ == "__main__":
    if len(sys.argv) != 3:
        print(__doc__)
        sys.exit(2)
    r = score_files(sys.argv[1], sys.argv[2])
    print(f"WER {r.wer:.4f}  ExpRate {r.exprate:.4f}  "
          f"<=1 {r.exprate_le1:.4f}  <=2 {r.exprate_le2:.4f}  (n={r.n_expressions})")
