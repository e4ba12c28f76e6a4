"""Shared leaves: the sharedgpu/* label contract (constants, labels),
the reference-format file logger, and MIOpen tuning wiring."""
