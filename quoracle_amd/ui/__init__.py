"""Web monitor + CLI tooling (the reference's Phoenix LiveView layer L7,
rebuilt as a thin FastAPI app over the event bus + store)."""
