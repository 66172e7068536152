"""Framework lint rules (reference: torchrec/linter/rules.py — TorchRec
ships AST lint rules like TR001; these are the MI355X-native equivalents).

Python rules (AST):
  TRA001  hidden device sync in the hot path: .item()/.cpu()/.numpy() inside
          torchrec_amd/distributed/{train_pipeline,comm_ops,dist_data}.py
          forward/backward code
  TRA002  bare `torch.cuda.synchronize()` outside tests/benchmarks

HIP rules (text, *.hip / csrc):
  TRH001  CUDA-compat shims: `#ifdef __CUDA`, `__CUDA_ARCH__`, hipify
          artifacts — this tree is CDNA4-native only
  TRH002  warp-32 idioms: a literal 32 used as a wave width (`& 31`,
          `>> 5` on lane math) — CDNA wavefronts are 64 wide; use kWaveSize
  TRH003  `__shfl_*_sync` (CUDA masked shuffles) — HIP shuffles take no mask

Run: python -m torchrec_amd.linter.rules [paths...]
"""

from __future__ import annotations

import ast
import re
import sys
from dataclasses import dataclass
from pathlib import Path
from typing import Iterator, List

SYNC_SENSITIVE = ("train_pipeline.py", "comm_ops.py", "dist_data.py")


@dataclass
class LintError:
    rule: str
    path: str
    line: int
    message: str

    def __str__(self) -> str:
        return f"{self.path}:{self.line}: {self.rule} {self.message}"


def _lint_python(path: Path) -> Iterator[LintError]:
    try:
        tree = ast.parse(path.read_text())
    except SyntaxError as e:
        yield LintError("TRA000", str(path), e.lineno or 0, f"syntax error: {e.msg}")
        return
    hot = path.name in SYNC_SENSITIVE
    for node in ast.walk(tree):
        if isinstance(node, ast.Call) and isinstance(node.func, ast.Attribute):
            name = node.func.attr
            if hot and name in ("item", "numpy") and not _in_comment_waiver(path, node.lineno):
                yield LintError(
                    "TRA001", str(path), node.lineno,
                    f".{name}() forces a device sync in pipeline-critical code "
                    "(waive with  # lint: sync-ok)",
                )
            if name == "synchronize" and isinstance(node.func.value, ast.Attribute):
                if getattr(node.func.value, "attr", "") == "cuda" and not _in_comment_waiver(
                    path, node.lineno
                ):
                    yield LintError(
                        "TRA002", str(path), node.lineno,
                        "torch.cuda.synchronize() in library code "
                        "(waive with  # lint: sync-ok)",
                    )


def _in_comment_waiver(path: Path, line: int) -> bool:
    try:
        text = path.read_text().splitlines()[line - 1]
    except IndexError:
        return False
    return "lint: sync-ok" in text


_HIP_RULES = [
    ("TRH001", re.compile(r"#\s*ifdef\s+__CUDA|__CUDA_ARCH__|#include\s+<cuda"),
     "CUDA-compat shim — this tree is CDNA4-native HIP only"),
    ("TRH002", re.compile(r"&\s*31\b|threadIdx\.x\s*>>\s*5\b"),
     "warp-32 idiom — CDNA wavefronts are 64 lanes (use kWaveSize / & 63)"),
    ("TRH003", re.compile(r"__shfl_\w*_sync"),
     "CUDA masked shuffle — HIP __shfl_* takes no mask"),
]


def _lint_hip(path: Path) -> Iterator[LintError]:
    for i, line in enumerate(path.read_text().splitlines(), start=1):
        if "lint: wave-ok" in line or line.lstrip().startswith("//"):
            continue
        for rule, pat, msg in _HIP_RULES:
            if pat.search(line):
                yield LintError(rule, str(path), i, msg)


def lint_paths(paths: List[str]) -> List[LintError]:
    errors: List[LintError] = []
    for root in paths:
        p = Path(root)
        files = [p] if p.is_file() else sorted(p.rglob("*"))
        for f in files:
            if f.suffix == ".py" and "test" not in f.name and "benchmarks" not in str(f):
                errors.extend(_lint_python(f))
            elif (
                f.suffix in (".hip", ".h")
                and "csrc" in str(f)
                and not f.name.endswith("_hip.hip")  # hipify build artifacts
            ):
                errors.extend(_lint_hip(f))
    return errors


def main(argv: List[str]) -> int:
    paths = argv or ["torchrec_amd"]
    errors = lint_paths(paths)
    for e in errors:
        print(e)
    return 1 if errors else 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
