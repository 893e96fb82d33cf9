"""Utility modules (parity: reference utils/)."""
