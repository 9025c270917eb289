"""Generated-project templates.

Each module exposes functions returning machinery.File / machinery.Fragments
for one group of generated files, mirroring the reference's template
packages (internal/plugins/workload/v1/scaffolds/templates/**).
"""
