#!/usr/bin/env python3
"""Mutation-testing tier (reference analog: the reference's mutation-testing
CI tier). AST-based operator mutations over the core parsing/
matching modules; each mutant must be KILLED by the targeted test files.

    python tools/mutation_check.py            # full sweep, prints report
    python tools/mutation_check.py --limit 5  # quick spot-check

A mutant "survives" when the targeted tests still pass with the mutated
module on disk — i.e., a seam the suite does not pin down. The committed
report lives at profiles/mutation_report.txt.
"""

from __future__ import annotations

import argparse
import ast
import copy
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent

# module under test -> test files that must kill its mutants
# (round-2 sweep widened per VERDICT item 9: services + auth + plugin
# framework, not just the 4 parsing modules)
TARGETS = {
    "mcp_context_forge_amd/ops/dfa.py": ["tests/test_dfa.py"],
    "mcp_context_forge_amd/utils/jsonpath.py": ["tests/test_jsonpath.py"],
    "mcp_context_forge_amd/protocol/jsonrpc.py": ["tests/test_jsonrpc.py"],
    "mcp_context_forge_amd/plugins/toon.py": ["tests/test_plugins.py"],
    "mcp_context_forge_amd/auth/jwt.py": ["tests/test_security_fixes.py", "tests/test_auth_breadth.py"],
    "mcp_context_forge_amd/auth/rsa.py": ["tests/test_auth_breadth.py", "tests/test_mutation_kills.py"],
    "mcp_context_forge_amd/auth/oauth.py": ["tests/test_auth_breadth.py", "tests/test_mutation_kills.py"],
    "mcp_context_forge_amd/plugins/framework.py": ["tests/test_plugins.py"],
    "mcp_context_forge_amd/plugins/external.py": ["tests/test_external_plugins.py"],
    "mcp_context_forge_amd/services/leader.py": ["tests/test_leader_election.py", "tests/test_mutation_kills.py"],
    "mcp_context_forge_amd/services/gateway_service.py": ["tests/test_federation_lifecycle.py",
                                                          "tests/test_federation.py"],
    "mcp_context_forge_amd/services/sessions.py": ["tests/test_http_app.py"],
}

# operator swaps (classic mutmut set, trimmed to the ones that typecheck)
_CMP = {ast.Lt: ast.LtE, ast.LtE: ast.Lt, ast.Gt: ast.GtE, ast.GtE: ast.Gt,
        ast.Eq: ast.NotEq, ast.NotEq: ast.Eq}
_BIN = {ast.Add: ast.Sub, ast.Sub: ast.Add}


class _Enumerator(ast.NodeVisitor):
    """Collect mutation sites: (node kind, lineno, col)."""

    def __init__(self):
        self.sites = []

    def visit_Compare(self, node):
        for op in node.ops:
            if type(op) in _CMP:
                self.sites.append(("cmp", node.lineno, node.col_offset))
                break
        self.generic_visit(node)

    def visit_BinOp(self, node):
        if type(node.op) in _BIN:
            self.sites.append(("bin", node.lineno, node.col_offset))
        self.generic_visit(node)

    def visit_Constant(self, node):
        if node.value is True or node.value is False:
            self.sites.append(("bool", node.lineno, node.col_offset))
        self.generic_visit(node)


class _Mutator(ast.NodeTransformer):
    def __init__(self, site):
        self.site = site
        self.applied = False

    def _is_site(self, kind, node):
        return (not self.applied and kind == self.site[0]
                and node.lineno == self.site[1] and node.col_offset == self.site[2])

    def visit_Compare(self, node):
        self.generic_visit(node)
        if self._is_site("cmp", node):
            node.ops = [_CMP[type(op)]() if type(op) in _CMP else op for op in node.ops]
            self.applied = True
        return node

    def visit_BinOp(self, node):
        self.generic_visit(node)
        if self._is_site("bin", node):
            node.op = _BIN[type(node.op)]()
            self.applied = True
        return node

    def visit_Constant(self, node):
        if self._is_site("bool", node):
            self.applied = True
            return ast.copy_location(ast.Constant(value=not node.value), node)
        return node


def run_tests(test_files) -> bool:
    """True = tests pass (mutant SURVIVES)."""
    r = subprocess.run([sys.executable, "-m", "pytest", "-x", "-q", "--timeout", "120",
                        "-p", "no:cacheprovider", *test_files],
                       cwd=ROOT, capture_output=True, timeout=600)
    return r.returncode == 0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--limit", type=int, default=0, help="max mutants per module (0 = all)")
    ap.add_argument("--stride", type=int, default=1, help="test every Nth site")
    args = ap.parse_args()

    report = []
    total = killed = 0
    for mod, tests in TARGETS.items():
        path = ROOT / mod
        src = path.read_text()
        tree = ast.parse(src)
        enum = _Enumerator()
        enum.visit(tree)
        sites = enum.sites[:: args.stride]
        if args.limit:
            sites = sites[: args.limit]
        survivors = []
        for site in sites:
            mtree = _Mutator(site)
            mutated = mtree.visit(copy.deepcopy(tree))
            ast.fix_missing_locations(mutated)
            if not mtree.applied:
                continue
            try:
                code = ast.unparse(mutated)
            except Exception:
                continue
            total += 1
            path.write_text(code)
            try:
                alive = run_tests(tests)
            finally:
                path.write_text(src)   # always restore
            if alive:
                survivors.append(site)
            else:
                killed += 1
            print(f"[{mod}] {site}: {'SURVIVED' if alive else 'killed'}", flush=True)
        report.append((mod, len(sites), survivors))

    print()
    lines = [f"mutation sweep: {killed}/{total} mutants killed "
             f"({100.0 * killed / max(total, 1):.0f}% kill rate)"]
    for mod, n, survivors in report:
        lines.append(f"  {mod}: {n - len(survivors)}/{n} killed")
        for s in survivors:
            lines.append(f"    SURVIVOR: {s[0]} at line {s[1]}:{s[2]}")
    out = "\n".join(lines)
    print(out)
    (ROOT / "profiles/mutation_report.txt").write_text(out + "\n")


if __name__ == "__main__":
    main()
