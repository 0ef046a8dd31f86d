#!/usr/bin/env python3
"""Unused-import linter (golangci-lint analog for this repo): package
__init__ re-exports (marked noqa) are exempt."""
import ast
import os
import sys

bad = 0
for root, dirs, files in os.walk("cro_amd"):
    dirs[:] = [d for d in dirs if d != "__pycache__"]
    for f in files:
        if not f.endswith(".py") or f == "__init__.py":
            continue
        path = os.path.join(root, f)
        src = open(path).read()
        tree = ast.parse(src)
        imported = {}
        for node in ast.walk(tree):
            if isinstance(node, ast.Import):
                for a in node.names:
                    imported[(a.asname or a.name).split(".")[0]] = node.lineno
            elif isinstance(node, ast.ImportFrom):
                for a in node.names:
                    if a.name != "*":
                        imported[a.asname or a.name] = node.lineno
        names = {n.id for n in ast.walk(tree) if isinstance(n, ast.Name)}
        attrs = {n.attr for n in ast.walk(tree) if isinstance(n, ast.Attribute)}
        lines = src.splitlines()
        for name, line in imported.items():
            if "noqa" in lines[line - 1]:
                continue
            if name in ("annotations",):
                continue
            if name not in names and name not in attrs and f"{name}." not in src:
                print(f"{path}:{line}: unused import {name}")
                bad += 1
sys.exit(1 if bad else 0)
