"""The layered-architecture contract, machine-enforced.

Parity with the reference's import-linter contract
(/root/reference/pyproject.toml:78-128): higher layers may import lower,
never the reverse, and the core never imports adapters/engine/models (the
heavy technology edges stay quarantined).

Enforced by static AST analysis of the in-repo sources — no import-linter
dependency needed.
"""

from __future__ import annotations

import ast
from pathlib import Path

PKG = Path(__file__).resolve().parents[2] / "calfkit_amd"

#: bottom (0) to top; a module may import only strictly-lower layers
LAYERS: list[list[str]] = [
    ["errors"],
    ["messages", "streaming"],
    ["schemas"],
    ["context"],
    ["ports", "mcp"],
    ["model"],
    ["providers"],
    ["hooks", "lifecycle"],
    ["definitions"],
    ["_engine", "_carriage", "_projection", "_policy"],
    ["_capabilities"],
    ["_surface"],
    ["_drive"],
    ["_wire"],
    ["_hosting"],
    ["_processors"],
    ["worker"],
]

#: outside the layered core; the CORE must never import these
QUARANTINE = {"adapters", "engine", "models", "parallel", "cli", "client",
              "testing", "ops", "utils"}

#: sanctioned lazy exceptions (in-function imports are exempt automatically;
#: these are module-level edges that are deliberate)
ALLOWED_EXTRA: dict[str, set[str]] = {}


def _layer_of(name: str) -> int | None:
    for i, mods in enumerate(LAYERS):
        if name in mods:
            return i
    return None


def _module_name(path: Path) -> str:
    rel = path.relative_to(PKG)
    return rel.parts[0].removesuffix(".py")


def _toplevel_imports(path: Path) -> set[str]:
    """calfkit_amd-internal imports at MODULE level only (lazy ones exempt)."""
    tree = ast.parse(path.read_text())
    found: set[str] = set()
    for node in tree.body:  # module level only
        if isinstance(node, ast.ImportFrom) and node.level >= 1:
            # relative: from . import X / from .foo import Y / from ..bar ...
            if node.level == 1 and node.module:
                found.add(node.module.split(".")[0])
            elif node.level == 1 and not node.module:
                found.update(a.name.split(".")[0] for a in node.names)
            elif node.level >= 2:
                if node.module:
                    found.add(node.module.split(".")[0])
        elif isinstance(node, ast.ImportFrom) and node.module and \
                node.module.startswith("calfkit_amd."):
            found.add(node.module.split(".")[1])
    return found


def _iter_core_files():
    for mods in LAYERS:
        for m in mods:
            p = PKG / f"{m}.py"
            if p.exists():
                yield m, [p]
            else:
                yield m, list((PKG / m).rglob("*.py"))


def test_layers_import_downward_only():
    violations = []
    for mod, files in _iter_core_files():
        layer = _layer_of(mod)
        for f in files:
            for imp in _toplevel_imports(f):
                if imp == mod:
                    continue
                target = _layer_of(imp)
                if target is None:
                    continue  # handled by the quarantine test
                if target > layer:
                    violations.append(f"{f.relative_to(PKG)} (layer {layer}) "
                                      f"imports {imp} (layer {target})")
    assert not violations, "upward imports:\n" + "\n".join(violations)


def test_core_never_imports_adapters_or_engine():
    violations = []
    for mod, files in _iter_core_files():
        for f in files:
            for imp in _toplevel_imports(f):
                if imp in QUARANTINE and imp not in ALLOWED_EXTRA.get(mod, set()):
                    violations.append(f"{f.relative_to(PKG)} imports {imp}")
    assert not violations, \
        "core imports quarantined tech:\n" + "\n".join(violations)


def test_adapters_are_sole_importers_of_their_sdk():
    """One importer per heavy SDK (pyproject.toml:130-155 analog)."""
    # httpx: providers.py + HTTP adapters, matching the reference's
    # sole-importer family (/root/reference/pyproject.toml:130-155:
    # "httpx -> providers.py + adapters")
    sdk_owners = {"aiokafka": ("adapters/kafka",),
                  "httpx": ("providers.py", "adapters/openai.py",
                            "adapters/anthropic.py", "adapters/mcp.py"),
                  "mcp": ("adapters/mcp.py",)}
    for f in PKG.rglob("*.py"):
        text = f.read_text()
        tree = ast.parse(text)
        for node in ast.walk(tree):
            names = []
            if isinstance(node, ast.Import):
                names = [a.name.split(".")[0] for a in node.names]
            elif isinstance(node, ast.ImportFrom) and node.module and node.level == 0:
                names = [node.module.split(".")[0]]
            for n in names:
                if n in sdk_owners:
                    rel = str(f.relative_to(PKG))
                    assert any(o in rel for o in sdk_owners[n]), \
                        f"{rel} imports {n}; only {sdk_owners[n]} may"
