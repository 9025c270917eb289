"""Reference-template parity oracle.

Renders the reference's own Go ``text/template`` bodies (extracted from
/root/reference/internal/plugins/workload/v1/scaffolds/templates/**)
with the field values THIS repo's pipeline computes, post-processes the
result exactly like this repo post-processes its own output (format_go,
standing in for machinery's imports.Process), and byte-diffs it against
the generated tree.  This measures the BASELINE.json north star
("byte-equivalent generated operator source") directly.

Usage: ``python -m operator_builder_amd.parity <generated-tree>`` after
an init+create-api run, or via tests/test_reference_parity.py.
"""

from .oracle import oracle_render_workload, diff_report, FileDiff

__all__ = ["oracle_render_workload", "diff_report", "FileDiff"]
