"""Misc utilities for lightgbm_amd."""
