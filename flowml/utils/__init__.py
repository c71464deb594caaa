"""Auxiliary subsystems: logging, config, checkpointing, profiling."""
