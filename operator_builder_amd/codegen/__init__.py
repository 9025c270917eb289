"""YAML -> Go object source generation."""

from .objectgen import generate, generate_node, GenerateError

__all__ = ["generate", "generate_node", "GenerateError"]
