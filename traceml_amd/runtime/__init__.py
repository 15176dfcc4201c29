"""Per-rank runtime: settings, identity, lifecycle, sampler agent, executor."""
