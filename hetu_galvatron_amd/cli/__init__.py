"""CLI entry points: train / search / profile_hardware / profile_model."""
