#!/usr/bin/env python3
"""Lightweight lint: every glt_amd module must import cleanly (catches
syntax errors, bad imports, circular imports) without needing a GPU."""
import importlib
import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

failed = []
for p in sorted(pathlib.Path("glt_amd").rglob("*.py")):
    mod = ".".join(p.with_suffix("").parts)
    if mod.endswith(".__init__"):
        mod = mod[: -len(".__init__")]
    try:
        importlib.import_module(mod)
    except ImportError as e:  # env-gated optional deps are fine
        if "common_io" in str(e) or "vineyard" in str(e):
            continue
        failed.append((mod, repr(e)))
    except Exception as e:  # noqa: BLE001
        failed.append((mod, repr(e)))
for mod, err in failed:
    print(f"FAIL {mod}: {err}")
print(f"{len(failed)} failures")
sys.exit(1 if failed else 0)
