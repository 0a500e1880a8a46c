"""Bucketing helpers."""
