"""Meta-test for docs/E2E_MATRIX.md (VERDICT r1 item 8 'done' criterion): every
mapped test function must actually exist, and the gap fraction must stay below
20% of the reference's e2e case inventory."""
import os
import re

DOC = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "docs", "E2E_MATRIX.md")


def _rows():
    rows = []
    for line in open(DOC):
        m = re.match(r"\|\s*([A-Z]+\d*\w*)\s*\|[^|]*\|\s*([^|]+?)\s*\|\s*"
                     r"(ok|partial|gap)\s*\|", line)
        if m:
            rows.append((m.group(1), m.group(2), m.group(3)))
    return rows


def test_matrix_rows_parse():
    rows = _rows()
    assert len(rows) >= 80, f"matrix table lost rows ({len(rows)})"
    ids = [r[0] for r in rows]
    assert len(ids) == len(set(ids)), "duplicate test IDs"


def test_matrix_gap_fraction_below_20_percent():
    rows = _rows()
    gaps = [r for r in rows if r[2] == "gap"]
    frac = len(gaps) / len(rows)
    assert frac < 0.20, (
        f"{len(gaps)}/{len(rows)} = {frac:.0%} gaps (must be < 20%): "
        + ", ".join(r[0] for r in gaps))


def test_matrix_mapped_functions_exist():
    import ast
    rows = [r for r in _rows() if r[2] != "gap"]
    tests_dir = os.path.dirname(os.path.abspath(__file__))
    defined = {}  # file -> set of test function names

    def names_in(path):
        if path not in defined:
            tree = ast.parse(open(path).read())
            out = set()
            for node in ast.walk(tree):
                if isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef)):
                    out.add(node.name)
            defined[path] = out
        return defined[path]

    missing = []
    for tid, ref, _status in rows:
        fname, func = ref.split("::", 1)
        path = os.path.join(tests_dir, fname)
        if not os.path.exists(path) or func not in names_in(path):
            missing.append(f"{tid} -> {ref}")
    assert not missing, "matrix maps to nonexistent tests:\n  " + \
        "\n  ".join(missing)
