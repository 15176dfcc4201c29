"""Core measurement layer: event names, arming gates, dual-clock timing,
GPU timestamp backends (HIP event pool + CDNA4 s_memrealtime ring stamps),
and per-step memory watermarks."""
