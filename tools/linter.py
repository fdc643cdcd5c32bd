"""Repo lint helper (reference tools/linter.py runs autopep8 over the
tree; autopep8 is not in this image, so this checks what CI here cares
about: every python file parses, no tabs, and lines stay under the limit).
"""

import ast
import pathlib
import sys

MAX_LINE = 100
CHECK_DIRS = ["megatron_amd", "tasks", "tools", "weights_conversion",
              "tests", "."]


def lint():
    root = pathlib.Path(__file__).resolve().parent.parent
    seen = set()
    problems = []
    for sub in CHECK_DIRS:
        base = root / sub
        paths = base.glob("*.py") if sub == "." else base.rglob("*.py")
        for path in paths:
            if "__pycache__" in str(path) or path in seen:
                continue
            seen.add(path)
            src = path.read_text()
            try:
                ast.parse(src)
            except SyntaxError as e:
                problems.append(f"{path}: syntax error: {e}")
                continue
            for i, line in enumerate(src.splitlines(), 1):
                if "\t" in line:
                    problems.append(f"{path}:{i}: tab character")
                if len(line) > MAX_LINE:
                    problems.append(f"{path}:{i}: line > {MAX_LINE} chars")
    for p in problems:
        print(p)
    print(f"checked {len(seen)} files, {len(problems)} problems")
    return len(problems)


if __name__ == "__main__":
    sys.exit(1 if lint() else 0)
