"""Line-level similarity sweep of repo sources against the reference tree.

Guards the from-scratch contract: every repo .py is compared against every
reference .py with difflib SequenceMatcher over stripped non-empty lines, and
the best match above a threshold is reported. Run in CI-ish settings as
`python tools/similarity_check.py [--threshold 0.5]`; exits nonzero when any
file exceeds the threshold so regressions are caught locally.
"""
from __future__ import annotations

import argparse
import difflib
import pathlib
import sys

REPO = pathlib.Path(__file__).resolve().parent.parent
REFERENCE = pathlib.Path("/root/reference")


def norm_lines(path: pathlib.Path) -> list[str]:
    out = []
    try:
        text = path.read_text(errors="ignore")
    except OSError:
        return out
    for ln in text.splitlines():
        s = ln.strip()
        if s:
            out.append(s)
    return out


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--threshold", type=float, default=0.5)
    ap.add_argument("--files", nargs="*", default=None,
                    help="repo-relative .py files to check (default: perceiver_amd tree)")
    args = ap.parse_args()

    if args.files:
        repo_files = [REPO / f for f in args.files]
    else:
        repo_files = sorted((REPO / "perceiver_amd").rglob("*.py"))
    ref_files = sorted(p for p in REFERENCE.rglob("*.py") if "test" not in p.name)
    ref_lines = {p: norm_lines(p) for p in ref_files}

    worst = []
    for rf in repo_files:
        mine = norm_lines(rf)
        if len(mine) < 10:
            continue
        best, best_ref = 0.0, None
        for ref, rl in ref_lines.items():
            if not rl or abs(len(rl) - len(mine)) > max(len(rl), len(mine)) * 0.7:
                continue
            r = difflib.SequenceMatcher(a=mine, b=rl, autojunk=False).ratio()
            if r > best:
                best, best_ref = r, ref
        worst.append((best, rf.relative_to(REPO), best_ref))

    worst.sort(reverse=True)
    bad = 0
    for score, rf, ref in worst[:30]:
        flag = " <-- OVER" if score > args.threshold else ""
        if score > args.threshold:
            bad += 1
        print(f"{score:.2f}  {rf}  ~  {ref.relative_to(REFERENCE) if ref else '-'}{flag}")
    return 1 if bad else 0


if __name__ == "__main__":
    sys.exit(main())
