"""Command-line interface (reference: cmd/operator-builder + pkg/cli)."""

from .main import main

__all__ = ["main"]
