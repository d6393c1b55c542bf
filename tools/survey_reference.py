"""Automate SURVEY.md §B: structural extraction of /root/reference.

The reference mount was empty through round 1; when it appears, run

    python tools/survey_reference.py [/root/reference] > survey_raw.md

and use the output to rewrite SURVEY.md §§1-6 with real citations.
Read-only; never writes into the reference tree.
"""
import re
import sys
from collections import Counter
from pathlib import Path


def scan(root: Path):
    out = []
    files = [p for p in sorted(root.rglob("*")) if p.is_file()]
    out.append(f"# Raw survey of {root} ({len(files)} files)\n")
    if not files:
        out.append("EMPTY — mount still not populated.\n")
        return "\n".join(out)

    by_ext = Counter(p.suffix or "(none)" for p in files)
    out.append("## Files by extension")
    for ext, n in by_ext.most_common():
        loc = sum(len(p.read_text(errors="replace").splitlines())
                  for p in files if (p.suffix or "(none)") == ext
                  and p.stat().st_size < 4_000_000)
        out.append(f"* {ext}: {n} files, ~{loc} lines")

    out.append("\n## Layout (top 2 levels)")
    seen = set()
    for p in files:
        rel = p.relative_to(root)
        key = rel.parts[:2]
        if key not in seen:
            seen.add(key)
            out.append(f"* {'/'.join(key)}")

    pats = {
        "classes/defs": r"^\s*(class |def )\w+",
        "gym surface": r"def (step|reset)\(|observation_space|action_space",
        "checkpointing": r"\b(save|restore|checkpoint|state_dict|Saver)\b",
        "collectives": r"ncclAllReduce|all_reduce|allreduce|dist\.|mpi|gloo|horovod",
        "kernels": r"__global__|cudaLaunch|<<<",
        "entry points": r'if __name__ == .__main__.|argparse|gflags|click',
        "traces": r"philly|trace|poisson|arrival",
    }
    for title, pat in pats.items():
        out.append(f"\n## {title}  (`{pat}`)")
        rx = re.compile(pat)
        hits = 0
        for p in files:
            if p.stat().st_size > 4_000_000:
                continue
            try:
                for i, line in enumerate(
                        p.read_text(errors="replace").splitlines(), 1):
                    if rx.search(line):
                        out.append(f"* {p.relative_to(root)}:{i}: "
                                   f"{line.strip()[:110]}")
                        hits += 1
                        if hits > 400:
                            out.append("* ... (truncated at 400 hits)")
                            return "\n".join(out)
            except OSError:
                pass
        if hits == 0:
            out.append("* (no hits)")
    return "\n".join(out)


if __name__ == "__main__":
    root = Path(sys.argv[1] if len(sys.argv) > 1 else "/root/reference")
    print(scan(root))
