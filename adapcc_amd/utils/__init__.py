"""Utilities: metrics counters, gradient-noise-scale estimators."""
