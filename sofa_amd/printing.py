"""ANSI console helpers (behavioral parity with reference bin/sofa_print.py:18-48)."""

from __future__ import annotations

import sys

C_TITLE = "\033[1;34m"
C_ERROR = "\033[1;31m"
C_WARN = "\033[1;33m"
C_INFO = "\033[1;32m"
C_HINT = "\033[1;36m"
C_PROG = "\033[1;35m"
C_END = "\033[0m"

_verbose = False


def set_verbose(v: bool) -> None:
    global _verbose
    _verbose = v


def print_title(msg: str) -> None:
    print(f"\n{C_TITLE}==== {msg} ===={C_END}")


def print_error(msg: str) -> None:
    print(f"{C_ERROR}[ERROR] {msg}{C_END}", file=sys.stderr)


def print_warning(msg: str) -> None:
    if _verbose:
        print(f"{C_WARN}[WARNING] {msg}{C_END}")


def print_info(msg: str) -> None:
    if _verbose:
        print(f"{C_INFO}[INFO] {msg}{C_END}")


def print_hint(msg: str) -> None:
    print(f"{C_HINT}[HINT] {msg}{C_END}")


def print_progress(msg: str) -> None:
    print(f"{C_PROG}[PROGRESS] {msg}{C_END}")
