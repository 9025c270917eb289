"""YAML -> Go object source generation."""

from .objectgen import generate, GenerateError

__all__ = ["generate", "GenerateError"]
