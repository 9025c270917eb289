"""Workload domain model (reference: internal/workload/v1)."""
