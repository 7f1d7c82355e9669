"""Regenerate tests/goldens/conformance_matrix.json: every vendored VCR
cassette request translated through every applicable provider translator.

Run from the repo root after an intentional translator change:
    python scripts/regen_conformance.py
then review the diff — the goldens lock translator behavior, so every
changed line must be explainable by the change you made.
"""

from __future__ import annotations

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tests.test_conformance_cassettes import build_matrix  # noqa: E402

OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "tests", "goldens", "conformance_matrix.json")


def main():
    matrix = build_matrix()
    with open(OUT, "w", encoding="utf-8") as f:
        json.dump(matrix, f, indent=1, sort_keys=True)
        f.write("\n")
    print(f"wrote {len(matrix)} cases to {OUT}")


if __name__ == "__main__":
    main()
