"""Static check of Session.run call sites against registered Func
signatures (reference analysis/typecheck + cmd/slicetypecheck).

  python -m bigslice_amd.tools.typecheck FILE [FILE...]

Walks each file's AST, finds `X = bigslice_amd.func(builder)` /
`bs.func(builder)` bindings and `sess.run(X, args...)` calls, and checks
the argument count against the builder's signature.  Exit code 1 on any
finding.
"""

from __future__ import annotations

import ast
import sys
from typing import Dict, List, Optional, Tuple


def _func_of_call(call: ast.Call) -> Optional[ast.expr]:
    f = call.func
    if isinstance(f, ast.Attribute) and f.attr == "func":
        return call.args[0] if call.args else None
    if isinstance(f, ast.Name) and f.id == "func":
        return call.args[0] if call.args else None
    return None


def _arity(fn: ast.expr, tree: ast.Module) -> Optional[Tuple[int, bool]]:
    """(num_required_args, has_varargs) of a lambda/def referenced."""
    target = None
    if isinstance(fn, ast.Lambda):
        target = fn.args
    elif isinstance(fn, ast.Name):
        for node in ast.walk(tree):
            if isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef)) \
                    and node.name == fn.id:
                target = node.args
                break
    if target is None:
        return None
    required = len(target.args) - len(target.defaults)
    return required, target.vararg is not None


def check_file(path: str) -> List[str]:
    with open(path) as fp:
        tree = ast.parse(fp.read(), filename=path)
    findings: List[str] = []
    funcs: Dict[str, Tuple[int, bool]] = {}
    # pass 1: Func registrations
    for node in ast.walk(tree):
        if isinstance(node, ast.Assign) and isinstance(node.value,
                                                       ast.Call):
            builder = _func_of_call(node.value)
            if builder is None:
                continue
            arity = _arity(builder, tree)
            if arity is None:
                continue
            for tgt in node.targets:
                if isinstance(tgt, ast.Name):
                    funcs[tgt.id] = arity
    # pass 2: run/must call sites
    for node in ast.walk(tree):
        if not (isinstance(node, ast.Call)
                and isinstance(node.func, ast.Attribute)
                and node.func.attr in ("run", "must")):
            continue
        if not node.args:
            continue
        first = node.args[0]
        if not isinstance(first, ast.Name) or first.id not in funcs:
            continue
        required, varargs = funcs[first.id]
        given = len(node.args) - 1
        if any(isinstance(a, ast.Starred) for a in node.args):
            continue
        if given != required and not varargs:
            findings.append(
                f"{path}:{node.lineno}: {first.id} takes {required} "
                f"argument(s), Session.{node.func.attr} passes {given}")
    return findings


def main():
    args = [a for a in sys.argv[1:] if not a.startswith("-")]
    if not args or "-h" in sys.argv or "--help" in sys.argv:
        print(__doc__)
        sys.exit(2)
    findings: List[str] = []
    for path in args:
        try:
            findings.extend(check_file(path))
        except (OSError, SyntaxError) as e:
            findings.append(f"{path}: {e}")
    for f in findings:
        print(f)
    sys.exit(1 if findings else 0)


if __name__ == "__main__":
    main()
