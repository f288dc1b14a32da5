"""History server: collector sidecar, storage backends, event replay,
dashboard-compatible HTTP API."""
