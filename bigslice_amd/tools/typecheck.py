"""Static check of Session.run call sites against registered Func
signatures (reference analysis/typecheck + cmd/slicetypecheck).

  python -m bigslice_amd.tools.typecheck FILE [FILE...]

Walks each file's AST, finds `X = bigslice_amd.func(builder)` /
`bs.func(builder)` bindings and `sess.run(X, args...)` calls, and
checks the argument count against the builder's signature AND, where
the builder annotates parameters with simple types (int/str/float/
bool/bytes/list/dict/tuple), the types of literal call arguments
against those annotations (the reference analyzer's type check,
analysis/typecheck/typecheck.go:44-150, best-effort like it: calls
through non-identifiers and non-literal args are skipped).  Exit
code 1 on any finding.
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


_SIMPLE_TYPES = {"int", "str", "float", "bool", "bytes", "list",
                 "dict", "tuple"}


def _annotations(args: ast.arguments) -> List[Optional[str]]:
    """Simple-type annotation name per positional parameter (None when
    absent or not a recognized simple type)."""
    out = []
    for a in args.args:
        ann = a.annotation
        out.append(ann.id if isinstance(ann, ast.Name)
                   and ann.id in _SIMPLE_TYPES else None)
    return out


def _literal_type(node: ast.expr) -> Optional[str]:
    """Type name of a literal argument expression (None = unknown)."""
    if isinstance(node, ast.Constant):
        if node.value is None:
            return None  # None may satisfy Optional annotations
        return type(node.value).__name__
    if isinstance(node, (ast.List, ast.ListComp)):
        return "list"
    if isinstance(node, (ast.Dict, ast.DictComp)):
        return "dict"
    if isinstance(node, ast.Tuple):
        return "tuple"
    if isinstance(node, ast.UnaryOp) and \
            isinstance(node.op, (ast.USub, ast.UAdd)):
        return _literal_type(node.operand)
    if isinstance(node, ast.JoinedStr):
        return "str"
    return None


def _compatible(given: str, annotated: str) -> bool:
    if given == annotated:
        return True
    # int literals satisfy float parameters (numeric widening); bools
    # are ints in Python
    if annotated == "float" and given in ("int", "bool"):
        return True
    if annotated == "int" and given == "bool":
        return True
    return False


def _arity(fn: ast.expr, tree: ast.Module
           ) -> Optional[Tuple[int, bool, List[Optional[str]]]]:
    """(num_required_args, has_varargs, annotations) of a referenced
    lambda/def."""
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
    return required, target.vararg is not None, _annotations(target)


def check_file(path: str) -> List[str]:
    with open(path) as fp:
        tree = ast.parse(fp.read(), filename=path)
    findings: List[str] = []
    funcs: Dict[str, Tuple[int, bool, List[Optional[str]]]] = {}
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
        required, varargs, anns = funcs[first.id]
        given = len(node.args) - 1
        if any(isinstance(a, ast.Starred) for a in node.args):
            continue
        if given != required and not varargs:
            findings.append(
                f"{path}:{node.lineno}: {first.id} takes {required} "
                f"argument(s), Session.{node.func.attr} passes {given}")
            continue
        # literal argument types vs the builder's simple annotations
        for i, arg in enumerate(node.args[1:]):
            if i >= len(anns) or anns[i] is None:
                continue
            lit = _literal_type(arg)
            if lit is not None and not _compatible(lit, anns[i]):
                findings.append(
                    f"{path}:{node.lineno}: {first.id} argument "
                    f"{i + 1} is {lit}, builder annotates {anns[i]}")
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
