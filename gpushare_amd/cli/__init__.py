"""Command-line entry points: daemon, inspect, podgetter."""
