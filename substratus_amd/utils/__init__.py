"""Shared utilities (GEMM solution tables, ...)."""
